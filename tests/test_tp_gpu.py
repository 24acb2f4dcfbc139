"""tp=2 on real GPUs over RCCL, with hipGraph-captured decode
(collectives in-graph). Runs whenever >=2 GPUs are visible; skipped on
1-GPU boxes (the driver's round-end tier is 1-GPU — the 8-GPU scaling
tier exercises this). VERDICT.md #6."""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_tp2_engine_rccl_graphs(tmp_path):
    if not torch.cuda.is_available() or torch.cuda.device_count() < 2:
        pytest.skip("needs >= 2 GPUs")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr", "127.0.0.1",
         "--master-port", "29617",
         str(Path(__file__).parent / "tp_gpu_worker.py"), str(tmp_path)],
        capture_output=True, text=True, timeout=600, env=env,
    )
    assert r.returncode == 0, f"worker failed:\n{r.stdout}\n{r.stderr}"
    assert "TP2_GPU_OK" in r.stdout
