"""Single-node multi-GPU runtime: one process per GPU over RCCL/xGMI.

This replaces the reference's Parsl HighThroughputExecutor task farm
(reference distllm/parsl.py:29-260): instead of pickling tasks to
share-nothing workers over ZMQ, the input file list is sharded across
ranks of a torch.distributed process group (backend "nccl" == RCCL on
ROCm, "gloo" on CPU test boxes), weights are broadcast once over xGMI,
and each rank writes its own HF-dataset shard (preserving the
reference's uuid4-dir + offline-merge data plane).

Launch: python -m torch.distributed.run --nproc-per-node N
        --master-addr 127.0.0.1 <script> ...
Single-process (no env) use works without init.
"""

from __future__ import annotations

import os
from typing import Literal, Sequence, TypeVar

import torch
import torch.distributed as dist

from ..utils import BaseConfig

T = TypeVar("T")


class ComputeConfig(BaseConfig):
    """Replaces the reference's ComputeConfigs (Parsl platforms).

    name "local": current process only (reference LocalConfig);
    name "node":  one rank per GPU via torch.distributed (reference
    WorkstationConfig's one-worker-per-GPU pinning, minus Parsl).
    The reference's HPC platform names (reference parsl.py: polaris =
    PBSPro, leonardo = Slurm) are accepted drop-in: in-process they
    behave like "node" (torchrun provides the ranks), and
    to_launch_config() maps the scheduler fields onto the batch-script
    renderer (`distllm launch`).
    """

    name: Literal["local", "node", "workstation", "slurm", "pbs",
                  "polaris", "leonardo"] = "local"
    ranks_per_node: int = 8
    backend: str = ""  # "" -> nccl if cuda else gloo
    # reference HPC platform fields (parsl.py:106-250)
    num_nodes: int = 1
    account: str = ""
    queue: str = ""
    walltime: str = "01:00:00"
    worker_init: str = ""
    scheduler_options: str = ""

    def to_launch_config(self):
        """HpcLaunchConfig for batch-script rendering, or None for the
        in-process platforms."""
        sched = {"slurm": "slurm", "leonardo": "slurm",
                 "pbs": "pbs", "polaris": "pbs"}.get(self.name)
        if sched is None:
            return None
        from .launcher import HpcLaunchConfig

        return HpcLaunchConfig(
            scheduler=sched, account=self.account, queue=self.queue,
            walltime=self.walltime, num_nodes=self.num_nodes,
            gpus_per_node=self.ranks_per_node,
            worker_init=self.worker_init,
            scheduler_options=self.scheduler_options,
        )


def init_distributed(backend: str = "") -> tuple[int, int]:
    """Initialize from torchrun env if present. Returns (rank, world)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    if not dist.is_initialized():
        if not backend:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(backend=backend)
    return dist.get_rank(), dist.get_world_size()


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


def shard_list(items: Sequence[T], rank: int | None = None,
               world: int | None = None) -> list[T]:
    """Strided (round-robin) shard of the input list for this rank —
    balances sorted-by-name inputs whose sizes correlate with position
    (file-level data parallelism; replaces ParslPoolExecutor.map
    fan-out)."""
    r = get_rank() if rank is None else rank
    w = get_world_size() if world is None else world
    return list(items[r::w])


@torch.no_grad()
def broadcast_model(model: torch.nn.Module, src: int = 0) -> None:
    """Broadcast weights from rank src over RCCL/xGMI. Bucketize small
    params into one flat tensor per dtype to avoid per-tensor launch +
    ring latency (xGMI links are p2p; few large messages >> many small).

    Note: the flat bucket transiently doubles the weights' memory
    footprint during broadcast — a deliberate trade on 288 GB HBM
    (even llama3-70b at 140 GB leaves ample headroom); switch to
    chunked buckets if a model ever crowds the device."""
    if not dist.is_initialized():
        return
    params = [p for p in model.parameters()] + \
             [b for b in model.buffers()]
    by_dtype: dict[torch.dtype, list[torch.Tensor]] = {}
    for p in params:
        by_dtype.setdefault(p.dtype, []).append(p)
    for dtype, group in by_dtype.items():
        flat = torch.cat([p.detach().reshape(-1) for p in group])
        dist.broadcast(flat, src=src)
        off = 0
        for p in group:
            n = p.numel()
            p.copy_(flat[off : off + n].view_as(p))
            off += n


def all_gather_object(obj):
    if not dist.is_initialized():
        return [obj]
    out = [None] * dist.get_world_size()
    dist.all_gather_object(out, obj)
    return out
