"""Multi-process (gloo, world_size=2) tests of the RCCL-runtime code
paths: file sharding, weight broadcast, bench contract."""

import json
import os
import subprocess
import sys
from pathlib import Path

import yaml

REPO = Path(__file__).parent.parent


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _torchrun(args, timeout=300):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               # these are gloo CPU-logic tests: hide any GPUs so
               # world_size 2 also runs on a 1-GPU box
               HIP_VISIBLE_DEVICES="", CUDA_VISIBLE_DEVICES="")
    # fresh port per call: a fixed port flakes on TIME_WAIT reuse
    return subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), *args],
        capture_output=True, text=True, timeout=timeout, env=env,
        cwd=REPO,
    )


def test_distributed_embedding_two_ranks(tmp_path):
    idir = tmp_path / "in"
    idir.mkdir()
    for i in range(4):
        with open(idir / f"f{i}.jsonl", "w") as fh:
            for j in range(3):
                fh.write(json.dumps({"text": f"Doc {i} line {j}.",
                                     "path": f"d{i}"}) + "\n")
    odir = tmp_path / "out"
    cfg = {
        "input_dir": str(idir),
        "output_dir": str(odir),
        "glob_patterns": ["*.jsonl"],
        "dataset_config": {"name": "jsonl", "batch_size": 2,
                           "num_data_workers": 0, "pin_memory": False},
        "encoder_config": {"name": "auto",
                           "pretrained_model_name_or_path": "bert-tiny",
                           "half_precision": False},
        "pooler_config": {"name": "mean"},
        "embedder_config": {"name": "full_sequence"},
        "writer_config": {"name": "huggingface"},
    }
    cfg_path = tmp_path / "cfg.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    r = _torchrun(["-m", "distllm_amd.distributed_embedding",
                   "--config", str(cfg_path)])
    assert r.returncode == 0, r.stdout + r.stderr
    shards = list((odir / "embeddings").iterdir())
    assert len(shards) == 4  # every file embedded by exactly one rank


def test_bench_two_ranks_gloo(tmp_path):
    r = _torchrun(["bench.py", "--gpus", "2", "--steps", "2",
                   "--warmup", "1", "--batch-size", "4",
                   "--seq-len", "32", "--model", "bert-tiny"])
    assert r.returncode == 0, r.stdout + r.stderr
    line = [ln for ln in r.stdout.splitlines()
            if ln.startswith("{") and "seqs/sec" in ln]
    assert len(line) == 1, r.stdout
    obj = json.loads(line[0])
    assert obj["n_gpus"] == 2
    assert obj["config"]["parallelism"] == "dp2"
    assert obj["config"]["global_batch"] == 8


def test_broadcast_model_helper():
    """broadcast_model on a single (non-initialized) process is a no-op
    and keeps weights intact."""
    import torch

    from distllm_amd.models import build_model
    from distllm_amd.parallel import broadcast_model

    m = build_model("bert-tiny", dtype=torch.float32)
    w0 = m.word_emb.clone()
    broadcast_model(m)
    assert torch.equal(w0, m.word_emb)


def test_tensor_parallel_two_ranks(tmp_path):
    """TP=2 decoder forward + engine decode match the single-rank model
    exactly (fp32, gloo)."""
    r = _torchrun(["tests/tp_worker.py", str(tmp_path)], timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "TP_OK" in r.stdout, r.stdout


def test_distributed_generation_two_ranks(tmp_path):
    idir = tmp_path / "in"
    idir.mkdir()
    for i in range(2):
        with open(idir / f"f{i}.jsonl", "w") as fh:
            for j in range(2):
                fh.write(json.dumps({"text": f"Prompt {i} {j} here",
                                     "path": f"p{i}"}) + "\n")
    odir = tmp_path / "out"
    cfg = {
        "input_dir": str(idir),
        "output_dir": str(odir),
        "glob_patterns": ["*.jsonl"],
        "prompt_config": {"name": "identity"},
        "reader_config": {"name": "jsonl"},
        "writer_config": {"name": "huggingface"},
        "generator_config": {"name": "native", "model": "llama-tiny",
                             "temperature": 0.0, "min_p": 0.0,
                             "max_tokens": 3, "max_model_len": 128,
                             "max_batch_size": 2},
    }
    cfg_path = tmp_path / "g.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    r = _torchrun(["-m", "distllm_amd.distributed_generation",
                   "--config", str(cfg_path)])
    assert r.returncode == 0, r.stdout + r.stderr
    shards = list((odir / "generations").iterdir())
    assert len(shards) == 2
    from datasets import load_from_disk

    total = sum(len(load_from_disk(str(s))) for s in shards)
    assert total == 4


def test_hpc_batch_script_rendering(tmp_path):
    """Slurm and PBS scripts carry the scheduler directives and a
    c10d-rendezvous torchrun line (reference Parsl platform parity)."""
    from distllm_amd.parallel.launcher import (
        HpcLaunchConfig,
        render_batch_script,
    )

    cmd = "-m distllm_amd.distributed_embedding --config e.yaml"
    s = render_batch_script(HpcLaunchConfig(
        scheduler="slurm", account="proj1", queue="gpu", num_nodes=4,
        walltime="02:00:00", scheduler_options="#SBATCH --exclusive",
        worker_init="module load rocm"), cmd)
    assert "#SBATCH --account=proj1" in s
    assert "#SBATCH --partition=gpu" in s
    assert "#SBATCH --nodes=4" in s
    assert "#SBATCH --exclusive" in s
    assert "module load rocm" in s
    assert "--nnodes=4 --nproc-per-node=8" in s
    assert "--rdzv-backend=c10d" in s
    assert "scontrol show hostnames" in s
    assert "HSA_ENABLE_IPC_MODE_LEGACY=0" in s
    assert s.startswith("#!/bin/bash")

    p = render_batch_script(HpcLaunchConfig(
        scheduler="pbs", account="proj2", queue="prod", num_nodes=2,
        gpus_per_node=4), cmd)
    assert "#PBS -A proj2" in p and "#PBS -q prod" in p
    assert "#PBS -l select=2:ngpus=4" in p
    assert "mpiexec -n 2 -ppn 1" in p
    assert "--nproc-per-node=4" in p
    assert 'head -n1 "$PBS_NODEFILE"' in p

    # CLI round-trip through YAML
    import yaml

    cfgp = tmp_path / "hpc.yaml"
    cfgp.write_text(yaml.safe_dump(
        {"scheduler": "slurm", "account": "a", "num_nodes": 2}))
    from typer.testing import CliRunner

    from distllm_amd.cli import app

    out = tmp_path / "job.sh"
    r = CliRunner().invoke(app, [
        "launch", "--config", str(cfgp), "--command", cmd,
        "--output", str(out)])
    assert r.exit_code == 0, r.output
    assert "#SBATCH --account=a" in out.read_text()


def test_reference_platform_compute_configs(tmp_path):
    """Reference-shaped compute_config blocks (parsl.py platform
    fields) parse drop-in and map onto the batch-script renderer."""
    import yaml

    from distllm_amd.parallel.launcher import render_batch_script
    from distllm_amd.parallel.runtime import ComputeConfig

    # polaris (PBSPro) shape from reference examples/embed/*.yaml
    cc = ComputeConfig(
        name="polaris", num_nodes=16,
        worker_init="module load rocm",
        scheduler_options="#PBS -l filesystems=home:eagle:grand",
        account="FoundEpidem", queue="demand", walltime="01:00:00")
    lc = cc.to_launch_config()
    s = render_batch_script(lc, "-m distllm_amd.distributed_embedding")
    assert "#PBS -A FoundEpidem" in s and "#PBS -q demand" in s
    assert "#PBS -l filesystems=home:eagle:grand" in s
    assert "select=16" in s and "module load rocm" in s

    # leonardo (Slurm) maps to sbatch
    lc2 = ComputeConfig(name="leonardo", account="x",
                        walltime="00:30:00").to_launch_config()
    assert lc2.scheduler == "slurm"
    # in-process platforms render nothing
    assert ComputeConfig(name="node").to_launch_config() is None

    # `distllm launch` consumes a full pipeline YAML's compute_config
    from typer.testing import CliRunner

    from distllm_amd.cli import app

    p = tmp_path / "embed.yaml"
    p.write_text(yaml.safe_dump({
        "input_dir": "/x", "output_dir": "/y",
        "dataset_config": {"name": "jsonl"},
        "encoder_config": {"name": "auto"},
        "pooler_config": {"name": "mean"},
        "embedder_config": {"name": "full_sequence"},
        "writer_config": {"name": "huggingface"},
        "compute_config": {"name": "polaris", "account": "proj",
                           "queue": "prod", "num_nodes": 2},
    }))
    out = tmp_path / "job.sh"
    r = CliRunner().invoke(app, [
        "launch", "--config", str(p),
        "--command", "-m distllm_amd.distributed_embedding "
        f"--config {p}", "--output", str(out)])
    assert r.exit_code == 0, r.output
    assert "#PBS -A proj" in out.read_text()
