"""Probe hipBLASLt GELU_AUX_BIAS epilogue support per (D dtype, aux dtype).

Run on a GPU box: python tools/hblt_probe.py
Prints the hipblas status for each combo and, where supported, the max
error of y (vs tanh-gelu reference) and aux (vs pre-activation).
"""

import sys

sys.path.insert(0, "/root/repo")

import torch

from skycomputing_amd.ops import hiplib
from skycomputing_amd.ops.functions import _dt, _stream, ptr


def main():
    lib = hiplib.require()
    torch.manual_seed(0)
    M, N, K = 256, 4096, 1024
    for dt in (torch.float32, torch.bfloat16):
        x = torch.randn(M, K, dtype=dt, device="cuda")
        w = torch.randn(N, K, dtype=dt, device="cuda") * 0.03
        b = torch.randn(N, dtype=dt, device="cuda")
        pre_ref = torch.nn.functional.linear(x.float(), w.float(), b.float())
        y_ref = torch.nn.functional.gelu(pre_ref, approximate="tanh")
        for aux_dt in (dt, torch.float32):
            y = torch.zeros(M, N, dtype=dt, device="cuda")
            aux = torch.zeros(M, N, dtype=aux_dt, device="cuda")
            rc = lib.sky_hblt_linear_gelu_aux(
                _stream(), ptr(x), ptr(w), ptr(b), ptr(y), ptr(aux),
                M, N, K, _dt(x), _dt(aux)
            )
            if rc == 0:
                torch.cuda.synchronize()
                ey = (y.float() - y_ref).abs().max().item()
                ea = (aux.float() - pre_ref).abs().max().item()
                print(f"D={dt} aux={aux_dt}: rc=0 max|y-ref|={ey:.2e} max|aux-pre|={ea:.2e}")
            else:
                print(f"D={dt} aux={aux_dt}: rc={rc}")


if __name__ == "__main__":
    main()
