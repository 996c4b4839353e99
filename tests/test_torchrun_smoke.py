"""End-to-end rehearsal of the driver's SCALE command path: torchrun
--nproc-per-node N bench.py on CPU/gloo (VERDICT r1 #3c).  Validates the
rendezvous env handling, the distributed branch of bench.py, and the JSON
contract line, without any GPU."""

import json
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_torchrun_gloo_bench_smoke():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
           "--master-port", "29517",
           os.path.join(repo, "bench.py"), "--gpus", "4",
           "--steps", "2", "--warmup", "1"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=280,
                       cwd=repo)
    assert r.returncode == 0, f"torchrun failed:\n{r.stdout[-2000:]}\n{r.stderr[-2000:]}"
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 4
    assert rec["metric"] == "attn_tflops"
    assert rec["value"] > 0
