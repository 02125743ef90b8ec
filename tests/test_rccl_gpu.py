"""RCCL collectives on MI355X hardware (round-1 verdict item 1).

Launches 2 torchrun ranks that SHARE one GPU (rank % device_count), so a
1-GPU box executes the real nccl(=RCCL) communicator paths the multi-GPU
debate round uses: the fused consensus all-gather
and the TP=2 per-layer all-reduce (C3). Requires
HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC), exported in this image.
"""

from __future__ import annotations

import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

WORKER = os.path.join(os.path.dirname(__file__), "rccl_worker.py")


def _torchrun(mode: str, port: int) -> subprocess.CompletedProcess:
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    return subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", str(port),
            WORKER, mode,
        ],
        capture_output=True, text=True, timeout=420, env=env,
    )


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_rccl_consensus_allgather_2ranks():
    r = _torchrun("consensus", 29411)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "RCCL_CONSENSUS_OK backend=nccl world=2" in r.stdout


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_rccl_tp2_parity():
    r = _torchrun("tp", 29412)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "RCCL_TP_OK backend=nccl world=2" in r.stdout
