"""End-to-end validation of the bench.py round flow (the driver contract).

Runs the REAL bench script under torchrun with world=2 on CPU/gloo
(--cpu-smoke): threaded co-resident opponents, deterministic per-opponent
consensus gather ordering across ranks, weak-scaling aggregate math.
This is the same code path the driver's 8-GPU RCCL run takes.
"""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_single_process_cpu_smoke():
    r = subprocess.run(
        [sys.executable, "bench.py", "--cpu-smoke", "--steps", "1",
         "--warmup", "0"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr[-800:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    assert d["n_gpus"] == 1 and d["config"]["opponents"] == 3
    assert d["value"] > 0 and d["higher_is_better"] is True
    assert d["scaling"] == "weak"


def test_torchrun_world2_gloo():
    env = dict(os.environ)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29537", "bench.py", "--cpu-smoke",
         "--steps", "1", "--warmup", "0"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-800:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    assert d["n_gpus"] == 2
    assert d["config"]["opponents"] == 6  # weak scaling: 3 per rank
    assert d["config"]["parallelism"] == "opponent-parallel dp2"
