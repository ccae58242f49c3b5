"""Client/parameter state management.

The reference shards machine-times over Ray CPU actors
(run_task.py:62-136); here per-client state lives in GPU memory.  The
global (server) model is ONE flat fp32 tensor with named views — so the
cross-GPU all-reduce of the aggregated delta is a single RCCL call on a
contiguous buffer (bucketing is a no-op at model sizes up to BERT-base),
and the fused update/aggregate kernels see one long contiguous vector
instead of dozens of small tensors.
"""

from __future__ import annotations

from typing import Dict, Iterator, List

import torch

from ..models.base import Params


class FlatParams:
    """A flat fp32 master vector with named per-parameter views."""

    def __init__(self, params: Params):
        self.shapes: Dict[str, tuple] = {k: tuple(v.shape) for k, v in params.items()}
        self.numels: Dict[str, int] = {k: v.numel() for k, v in params.items()}
        total = sum(self.numels.values())
        device = next(iter(params.values())).device
        self.flat = torch.empty(total, dtype=torch.float32, device=device)
        self.views: Params = {}
        offs = [0]
        off = 0
        for k, v in params.items():
            n = v.numel()
            view = self.flat[off:off + n].view(self.shapes[k])
            view.copy_(v.detach().float())
            self.views[k] = view
            off += n
            offs.append(off)
        # per-parameter block offsets (element index into `flat`); the
        # fused flat kernels use these to map a replica-buffer position
        # back to its global-master element
        self.offsets = torch.tensor(offs, dtype=torch.int64, device=device)

    @property
    def device(self) -> torch.device:
        return self.flat.device

    def numel(self) -> int:
        return self.flat.numel()

    def cast(self, dtype: torch.dtype) -> Params:
        """Per-parameter cast of the master (new tensors, e.g. bf16)."""
        return {k: v.to(dtype) for k, v in self.views.items()}

    def zeros_like_flat(self) -> torch.Tensor:
        return torch.zeros_like(self.flat)

    def views_of(self, flat: torch.Tensor) -> Params:
        """Named views into another flat tensor with this layout."""
        out: Params = {}
        off = 0
        for k, shape in self.shapes.items():
            n = self.numels[k]
            out[k] = flat[off:off + n].view(shape)
            off += n
        return out

    def state_dict(self) -> Params:
        return {k: v.detach().clone() for k, v in self.views.items()}

    def load_state_dict(self, sd: Params) -> None:
        for k, v in sd.items():
            self.views[k].copy_(v.float())


def replicate_params(cast_params: Params, clients: int) -> Params:
    """Per-parameter [C, ...] leaf replicas.

    Separate leaves (not views of one flat buffer) so autograd writes
    each gradient straight into its own tensor — a single flat leaf
    makes every view's backward materialise a full-size zeros + copy +
    accumulate chain, which measured ~50% of a ResNet round
    (profiles/resnet_round_r01.md).  The fused update/delta kernels
    take the tensor list (one small launch per parameter).
    """
    ops = None
    if cast_params and next(iter(cast_params.values())).is_cuda:
        from ..ops.fused import load_hip_ops
        ops = load_hip_ops()
    out: Params = {}
    for k, v in cast_params.items():
        vd = v.detach()
        if (ops is not None and vd.numel() % 8 == 0
                and vd.dtype in (torch.bfloat16, torch.float32)):
            # broadcast kernel (replicate.hip): torch's expand().clone()
            # runs the stride-0 source through an unvectorised copy
            # (~0.39 TB/s measured; ~65 ms/round on the flagship)
            rep = ops.replicate(vd.contiguous(), clients)
        else:
            # clone, not contiguous(): for clients==1 an expand of a
            # contiguous tensor is already contiguous and .contiguous()
            # would RETURN THE SAME STORAGE — the "replica" would alias
            # the global master and local training would corrupt it.
            rep = vd.unsqueeze(0).expand(clients, *v.shape) \
                    .clone(memory_format=torch.contiguous_format)
        rep.requires_grad_(True)
        out[k] = rep
    return out


def replicate_flat(cast_params: Params, clients: int) -> torch.Tensor:
    """Build ONE flat replica buffer holding every client's weights.

    Layout is *param-major*: a 1-D tensor of length C*P laid out as the
    concatenation over parameters of their [C, *shape] client-batched
    blocks.  Two things follow from this layout:
    - ``batched_views(buf)`` returns zero-copy contiguous views, so a
      forward pass costs no repack;
    - the buffer is the single autograd LEAF — gradients accumulate into
      one flat ``buf.grad`` with the *same* layout, so the fused
      SGD/FedProx update and the weighted delta reduction are each one
      kernel launch over contiguous memory, and the flat fp32 master
      (FlatParams.flat, param-major with the same dict order) lines up
      block for block.
    """
    sample = next(iter(cast_params.values()))
    total = sum(v.numel() for v in cast_params.values())
    buf = torch.empty(clients * total, dtype=sample.dtype, device=sample.device)
    off = 0
    for v in cast_params.values():
        n = v.numel()
        buf[off:off + clients * n].view(clients, *v.shape).copy_(
            v.detach().unsqueeze(0).expand(clients, *v.shape))
        off += clients * n
    buf.requires_grad_(True)
    return buf


def batched_views(buf: torch.Tensor, shapes: Dict[str, tuple],
                  clients: int) -> Params:
    """Differentiable named [C, ...] views into a replicate_flat buffer."""
    out: Params = {}
    off = 0
    for k, shape in shapes.items():
        n = 1
        for s in shape:
            n *= s
        out[k] = buf[off:off + clients * n].view(clients, *shape)
        off += clients * n
    return out


def chunk_ids(ids: List[int], chunk: int) -> Iterator[List[int]]:
    for i in range(0, len(ids), chunk):
        yield ids[i:i + chunk]
