"""The driver's scaling contract: bench.py under torch.distributed.run with
one rank per (virtual) GPU — here 2 ranks on CPU over gloo."""

import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent.parent


def test_bench_two_ranks_gloo():
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29571", "bench.py", "--gpus", "2", "--steps", "1",
           "--warmup", "0", "--gib-per-step", "1", "--jobs", "2",
           "--engine", "host"]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=400)
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, f"no JSON result:\n{out.stdout[-2000:]}\n{out.stderr[-2000:]}"
    d = json.loads(lines[-1])
    assert d["n_gpus"] == 2
    assert d["value"] > 0
    assert d["scaling"] == "weak"
