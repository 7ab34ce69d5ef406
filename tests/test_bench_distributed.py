"""The driver launches bench.py under torch.distributed.run with one
rank per GPU — keep that path covered on CPU (gloo, world_size 2)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_under_torchrun_gloo():
    env = dict(os.environ)
    env["CUDA_VISIBLE_DEVICES"] = ""  # force gloo even on a GPU box
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29597", "bench.py", "--gpus", "2",
         "--steps", "5", "--warmup", "1"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines()
            if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["metric"] == "p50 Allocate() RPC latency"
    assert d["n_gpus"] == 2 and d["steps"] == 5
    assert d["value"] > 0 and not d["higher_is_better"]
