"""Build the in-tree HIP kernel library for gfx950.

Usage: ``python -m skycomputing_amd.ops.build`` (also driven by
__graft_entry__.build()). Compiles every ops/hip/*.hip with hipcc
--offload-arch=gfx950 into ops/hip/libskyhip.so. The .so is git-ignored but
travels with gpurun snapshots, so GPU boxes load exactly what was built
here (hipcc cross-compiles without a GPU).
"""

from __future__ import annotations

import glob
import hashlib
import json
import os
import subprocess
import sys

HIP_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "hip")
LIB = os.path.join(HIP_DIR, "libskyhip.so")
STAMP = os.path.join(HIP_DIR, ".build_stamp.json")

HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("SKY_GFX_ARCH", "gfx950")
CFLAGS = [
    f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
    "-fvisibility=hidden", "-Wall", "-Wno-unused-function",
]


def _source_hash(sources: list[str]) -> str:
    h = hashlib.sha256()
    for s in sorted(sources) + [os.path.join(HIP_DIR, "common.h")]:
        with open(s, "rb") as f:
            h.update(f.read())
    h.update(" ".join(CFLAGS).encode())
    return h.hexdigest()


def build(force: bool = False, verbose: bool = True) -> str:
    sources = sorted(glob.glob(os.path.join(HIP_DIR, "*.hip")))
    if not sources:
        raise RuntimeError(f"no .hip sources under {HIP_DIR}")
    want = _source_hash(sources)
    if not force and os.path.isfile(LIB) and os.path.isfile(STAMP):
        try:
            with open(STAMP) as f:
                if json.load(f).get("hash") == want:
                    if verbose:
                        print(f"[skyhip] up to date: {LIB}")
                    return LIB
        except Exception:
            pass
    objs = []
    for src in sources:
        obj = src.replace(".hip", ".o")
        cmd = [HIPCC, *CFLAGS, "-x", "hip", "-c", src, "-o", obj]
        if verbose:
            print("[skyhip]", " ".join(cmd))
        subprocess.run(cmd, check=True)
        objs.append(obj)
    cmd = [HIPCC, f"--offload-arch={ARCH}", "-shared", "-fPIC", *objs,
           "-L/opt/rocm/lib", "-lhipblaslt", "-o", LIB]
    if verbose:
        print("[skyhip]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    with open(STAMP, "w") as f:
        json.dump({"hash": want}, f)
    if verbose:
        print(f"[skyhip] built {LIB}")
    return LIB


if __name__ == "__main__":
    build(force="--force" in sys.argv)
