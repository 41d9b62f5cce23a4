"""End-to-end demonstration of the headline capability: benchmark-driven
optimal allocation beats even allocation under injected heterogeneity
(the reference's 55% claim, measured wall-clock here on CPU/gloo)."""

from __future__ import annotations

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_optimal_beats_even_wallclock(tmp_path):
    out = tmp_path / "speedup.json"
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run", "--standalone",
            "--local-addr", "127.0.0.1", "--nproc-per-node", "3",
            os.path.join(REPO, "experiment", "speedup_bench.py"),
            "--layers", "4", "--batch", "8", "--seq", "16",
            "--hidden", "64", "--heads", "4",
            "--steps", "4", "--warmup", "1", "--microbatches", "2",
            "--slowdowns", "0,0,6", "--json-out", str(out),
        ],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stdout[-3000:] + r.stderr[-3000:]
    data = json.loads(out.read_text())
    ms = data["results_ms"]
    speedup = data["speedup_optimal_vs_even"]
    # rank 2 is 6x slowed: even allocation bottlenecks on it, optimal
    # shifts layers away -> meaningful wall-clock speedup. The margin is
    # theoretical ~2x; accept >1.1 to tolerate loaded-CI noise.
    assert speedup > 1.1, (ms, speedup)
