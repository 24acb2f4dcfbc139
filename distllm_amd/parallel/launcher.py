"""HPC batch-script generation for multi-node launches.

Reference parity: the reference's Parsl platform configs submit to
Slurm (LeonardoSettings, distllm/parsl.py:106-170) and PBSPro
(PolarisConfig, parsl.py:172-250) with account/queue/walltime/
scheduler_options fields. This framework's scale-out unit is
`torch.distributed.run` (one rank per GPU over RCCL), so the platform
layer reduces to *rendering the batch script* that brings torchrun up
on every node — no task-farm middleware to install on the cluster.

Rendezvous uses the c10d backend against the first allocated host, so
no node-rank bookkeeping is needed on either scheduler.
"""

from __future__ import annotations

from typing import Literal

from ..utils import BaseConfig


class HpcLaunchConfig(BaseConfig):
    """Fields mirror the reference's Slurm/PBS platform configs."""

    scheduler: Literal["slurm", "pbs"] = "slurm"
    account: str = ""
    queue: str = ""                  # Slurm partition / PBS queue
    walltime: str = "01:00:00"
    num_nodes: int = 1
    gpus_per_node: int = 8
    job_name: str = "distllm"
    scheduler_options: str = ""      # extra raw #SBATCH / #PBS lines
    worker_init: str = ""            # e.g. "module load rocm"
    master_port: int = 29500


def render_batch_script(cfg: HpcLaunchConfig, command: str) -> str:
    """Render a submit-ready batch script that runs `command` (a
    `python -m distllm_amd.distributed_embedding ...`-style module
    invocation) under torchrun on every allocated node."""
    torchrun = (
        "python -m torch.distributed.run "
        f"--nnodes={cfg.num_nodes} "
        f"--nproc-per-node={cfg.gpus_per_node} "
        "--rdzv-backend=c10d "
        f'--rdzv-endpoint="$MASTER_ADDR":{cfg.master_port} '
        '--rdzv-id="$JOB_ID" '
        f"{command}"
    )
    common = [
        "",
        "# RCCL over the host driver's dmabuf IPC path",
        "export HSA_ENABLE_IPC_MODE_LEGACY=0",
    ]
    if cfg.worker_init:
        common += ["", cfg.worker_init]
    if cfg.scheduler == "slurm":
        lines = [
            "#!/bin/bash",
            f"#SBATCH --job-name={cfg.job_name}",
            f"#SBATCH --nodes={cfg.num_nodes}",
            "#SBATCH --ntasks-per-node=1",
            f"#SBATCH --gres=gpu:{cfg.gpus_per_node}",
            f"#SBATCH --time={cfg.walltime}",
        ]
        if cfg.account:
            lines.append(f"#SBATCH --account={cfg.account}")
        if cfg.queue:
            lines.append(f"#SBATCH --partition={cfg.queue}")
        if cfg.scheduler_options:
            lines += cfg.scheduler_options.splitlines()
        lines += common + [
            "",
            'MASTER_ADDR=$(scontrol show hostnames '
            '"$SLURM_JOB_NODELIST" | head -n1)',
            'JOB_ID="$SLURM_JOB_ID"',
            f"srun --ntasks-per-node=1 {torchrun}",
            "",
        ]
    else:  # pbs
        lines = [
            "#!/bin/bash",
            f"#PBS -N {cfg.job_name}",
            f"#PBS -l select={cfg.num_nodes}:ngpus={cfg.gpus_per_node}",
            f"#PBS -l walltime={cfg.walltime}",
        ]
        if cfg.account:
            lines.append(f"#PBS -A {cfg.account}")
        if cfg.queue:
            lines.append(f"#PBS -q {cfg.queue}")
        if cfg.scheduler_options:
            lines += cfg.scheduler_options.splitlines()
        lines += common + [
            "",
            'cd "$PBS_O_WORKDIR"',
            'MASTER_ADDR=$(head -n1 "$PBS_NODEFILE")',
            'JOB_ID="$PBS_JOBID"',
            f"mpiexec -n {cfg.num_nodes} -ppn 1 {torchrun}",
            "",
        ]
    return "\n".join(lines)
