"""RCCL/xGMI distributed layer.

The reference has no collectives at all — its "distributed backend" is
gRPC + Pulsar between services.  Here scale-out is one
process per GPU: ``torch.distributed`` with backend "nccl" (which IS
RCCL on ROCm) over the node's xGMI links, "gloo" on CPU for tests.

FedAvg needs exactly two collective moments per round:
  - all_reduce(SUM) of the flat fp32 delta accumulator (one contiguous
    buffer, engine/client_manager.FlatParams layout) + of the scalar
    total weight — after which every rank applies the same update, so no
    weight broadcast is ever needed after init;
  - an initial broadcast of the master so all ranks start identical.

xGMI is 7 point-to-point links per GPU; a single big contiguous
all-reduce (ring/tree chosen by RCCL) at model size (MLP 159 KB …
BERT-base 440 MB fp32) is the right granularity — no bucketing split is
needed because there is only ONE buffer.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    device: str = "cpu"
    backend: str = ""

    @property
    def enabled(self) -> bool:
        return self.world_size > 1


def init_distributed(device: Optional[str] = None,
                     timeout_s: int = 600) -> DistContext:
    """Initialise from torchrun env vars; single-process no-op otherwise."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_gpu = torch.cuda.is_available()
    if device is None:
        device = f"cuda:{local_rank}" if use_gpu else "cpu"
    if world <= 1:
        return DistContext(rank=0, world_size=1, local_rank=0, device=device)
    backend = "nccl" if use_gpu else "gloo"
    if use_gpu:
        torch.cuda.set_device(local_rank)
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s))
    return DistContext(rank=rank, world_size=world, local_rank=local_rank,
                       device=device, backend=backend)


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def all_reduce_flat(flat: torch.Tensor, async_op: bool = False):
    """Sum-all-reduce one contiguous buffer across ranks (RCCL on GPU)."""
    if not is_distributed():
        return None
    return dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=async_op)


def broadcast_flat(flat: torch.Tensor, src: int = 0) -> None:
    if is_distributed():
        dist.broadcast(flat, src=src)


def barrier() -> None:
    if is_distributed():
        dist.barrier()
