"""Driver-contract guard: bench.py must emit exactly one JSON line with the
required schema from a plain single-process invocation."""

from __future__ import annotations

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract(tmp_path):
    out = tmp_path / "b.json"
    env = dict(os.environ)
    env["MASTER_PORT"] = str(21000 + os.getpid() % 20000)
    r = subprocess.run(
        [sys.executable, "bench.py", "--layers", "1", "--batch", "8",
         "--seq", "16", "--steps", "2", "--warmup", "1",
         "--json-out", str(out)],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=420,
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1
    d = json.loads(json_lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["higher_is_better"] is False
    assert d["scaling"] == "strong"
    assert d["data"] == "synthetic"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and abs(d["ms_per_step"] - d["value"] * 1e3) < 1e-6
    cfg = d["config"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in cfg, key
    assert json.loads(out.read_text()) == d


def _torchrun_bench(nproc, extra, out, attempts=3):
    """Run bench.py under torch.distributed.run; one retry on a fresh
    rendezvous port (loaded machines occasionally drop the first
    rendezvous or overrun the startup window)."""
    r = None
    for attempt in range(attempts):
        port = str(20000 + (os.getpid() * 13 + attempt * 101 + nproc) % 40000)
        env = dict(os.environ)
        r = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
             "--master-port", port, "bench.py", *extra,
             "--json-out", str(out)],
            cwd=REPO, env=env, capture_output=True, text=True, timeout=560,
        )
        if r.returncode == 0:
            return r
    return r


def test_bench_8rank_driver_invocation(tmp_path):
    """Exactly the command shape the driver uses for the 8-GPU scaling run
    (torch.distributed.run, one rank per device) — on gloo/CPU with a tiny
    model, so the full multi-rank bench path (partition, pipeline schedule,
    microbatching, JSON emission) is exercised before it ever meets 8 GPUs."""
    out = tmp_path / "b8.json"
    r = _torchrun_bench(8, ["--gpus", "8", "--layers", "4", "--batch", "8",
                            "--seq", "16", "--steps", "2", "--warmup", "1"], out)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, r.stdout[-2000:]
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "pp8"
    assert d["config"]["microbatches"] == 4  # default M = N/2 (measured W(M) curve)
    assert d["value"] > 0


def test_bench_2rank_default_microbatches(tmp_path):
    """N=2 with driver-default flags: default M = N/2 = 1 (sequential
    relay — the measured W(M) optimum at this batch)."""
    out = tmp_path / "b2.json"
    r = _torchrun_bench(2, ["--gpus", "2", "--layers", "4", "--batch", "8",
                            "--seq", "16", "--steps", "2", "--warmup", "1"], out)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    d = json.loads([l for l in r.stdout.splitlines() if l.startswith("{")][0])
    assert d["config"]["parallelism"] == "pp2"
    assert d["config"]["microbatches"] == 1
    assert d["value"] > 0


def test_bench_virtual_stages_cpu(tmp_path):
    """Opt-in interleaved virtual stages through the bench entrypoint
    (4 ranks x 2 chunks on gloo/CPU)."""
    out = tmp_path / "bv.json"
    r = _torchrun_bench(4, ["--gpus", "4", "--layers", "4", "--batch", "8",
                            "--seq", "16", "--steps", "2", "--warmup", "1",
                            "--virtual-stages", "2"], out)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    d = json.loads([l for l in r.stdout.splitlines() if l.startswith("{")][0])
    assert d["config"]["parallelism"] == "pp4x2"
    assert d["value"] > 0


def test_bench_virtual_stages_optimal_allocation(tmp_path):
    """--virtual-stages combined with benchmark-driven allocation routes
    through Allocator.interleaved_allocate (heterogeneity-aware chunk
    sizing) end to end."""
    out = tmp_path / "bvo.json"
    r = _torchrun_bench(2, ["--gpus", "2", "--layers", "4", "--batch", "8",
                            "--seq", "16", "--steps", "1", "--warmup", "1",
                            "--virtual-stages", "2", "--allocate", "optimal",
                            "--stimulate"], out)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    d = json.loads([l for l in r.stdout.splitlines() if l.startswith("{")][0])
    assert d["config"]["parallelism"] == "pp2x2"
    assert d["config"]["allocate"] == "optimal"
    assert d["value"] > 0
