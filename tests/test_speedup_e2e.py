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
    # wall-clock comparison: one re-measure on a loaded machine before
    # declaring failure (the margin is theoretical ~2x; threshold 1.1)
    last = None
    for attempt in range(2):
        out = tmp_path / f"speedup{attempt}.json"
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
        last = (data["results_ms"], data["speedup_optimal_vs_even"])
        if last[1] > 1.1:
            return
    assert last[1] > 1.1, last
