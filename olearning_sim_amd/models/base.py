"""Client-batched functional model base.

The reference contains no model code at all — each virtual phone runs a
user operator script in a subprocess (utils_run_task.py:496-514).  Here a
model family is a *functional* module whose parameters carry a leading
client dimension: params[name] has shape [C, ...] (one weight set per
co-resident virtual client), and forward consumes inputs [C, B, ...]
computing all C clients' local models in a handful of big batched GPU
ops.  This is what lets one MI355X simulate thousands of phones per
round instead of one subprocess each.

Conventions
- ``init_global(device, dtype, generator)`` -> dict of UN-batched
  parameter tensors (the global/server model; fp32 master).
- ``replicate(global_params, clients)`` -> dict of [C, ...] leaf tensors
  (requires_grad) each client starts the round from.
- ``forward(params, x)`` -> logits.  Conv stacks use the channel-grouped
  layout [B, C*ch, H, W] (one grouped conv per layer, groups=C) so no
  per-layer permutes are needed; dense stacks use [C, B, F] with bmm.
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import torch
import torch.nn.functional as F

Params = Dict[str, torch.Tensor]


class ClientBatchedModel:
    name: str = "base"
    num_classes: int = 10
    input_shape = (1,)          # per-sample shape (no batch dims)
    sequence_model: bool = False
    # rough activation elements held live per sample during fwd+bwd
    # (sizes the engine's auto chunking; override per model)
    act_elems_per_sample: int = 4096

    # -- parameter management -------------------------------------------
    def param_shapes(self) -> Dict[str, tuple]:
        raise NotImplementedError

    def init_global(self, device="cpu", dtype=torch.float32,
                    generator: Optional[torch.Generator] = None) -> Params:
        raise NotImplementedError

    @staticmethod
    def replicate(global_params: Params, clients: int,
                  dtype: torch.dtype) -> Params:
        """Clone the global model into [C, ...] per-client leaves."""
        out: Params = {}
        for k, v in global_params.items():
            # clone (not .contiguous()): at clients==1 the expand is
            # already contiguous and would alias the source storage
            rep = v.detach().to(dtype).unsqueeze(0).expand(
                clients, *v.shape).clone(memory_format=torch.contiguous_format)
            rep.requires_grad_(True)
            out[k] = rep
        return out

    def forward(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def loss(self, params: Params, x: torch.Tensor,
             y: torch.Tensor) -> torch.Tensor:
        """Mean CE over clients*batch. x: [C,B,...], y: [C,B]."""
        logits = self.forward(params, x)           # [C, B, K]
        n, k = logits.shape[0] * logits.shape[1], logits.shape[-1]
        from ..ops import cross_entropy_fwd_bwd
        return cross_entropy_fwd_bwd(logits.reshape(n, k), y.reshape(n))


# -- batched primitives ----------------------------------------------------

def binit(shape, fan_in: int, device, dtype, generator) -> torch.Tensor:
    """Kaiming-uniform init matching nn.Linear/Conv2d defaults."""
    bound = 1.0 / math.sqrt(fan_in) if fan_in > 0 else 0.0
    t = torch.empty(shape, device=device, dtype=torch.float32)
    t.uniform_(-bound, bound, generator=generator)
    return t.to(dtype)


def kaiming(shape, fan_in: int, device, dtype, generator) -> torch.Tensor:
    gain = math.sqrt(2.0)
    std = gain / math.sqrt(fan_in)
    bound = math.sqrt(3.0) * std
    t = torch.empty(shape, device=device, dtype=torch.float32)
    t.uniform_(-bound, bound, generator=generator)
    return t.to(dtype)


import os as _os

# hipBLASLt memory-faults on strided-transposed bmm operands at batch
# counts >= ~100 UNDER REAL WORKLOADS on this stack: isolated probes of
# every backward shape pass (tools/bmmprobe.py, maxerr 0.0), but the
# full BERT-base bench at C=125 faults/hangs with view operands and
# runs clean with materialised transposes (A/B: gpurun_out/
# b_bert6safe.log ok vs b_bert6view.log rc=124) — the trigger is
# allocator/workspace state, not the shape.  Default stays SAFE
# (materialise); OLSIM_BMM_SAFE=0 opts into views for future stacks.
_BMM_SAFE = _os.environ.get("OLSIM_BMM_SAFE", "1") != "0"


class _BLinearFn(torch.autograd.Function):
    """bmm-based per-client linear with an explicit backward (the
    transposed backward operands stay VIEWS — no copy traffic; the
    BERT-base profile showed materialised transposes were the dominant
    elementwise cost, profiles/bert_full_logits_r02.md)."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        y = torch.bmm(x, w)
        if b is not None:
            y = y + b.unsqueeze(1)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if _BMM_SAFE:
                from ..ops.fused import fast_transpose
                dx = torch.bmm(dy, fast_transpose(w))
            else:
                dx = torch.bmm(dy, w.transpose(1, 2))
        if ctx.needs_input_grad[1]:
            if _BMM_SAFE:
                from ..ops.fused import fast_transpose
                dw = torch.bmm(fast_transpose(x), dy)
            else:
                dw = torch.bmm(x.transpose(1, 2), dy)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = dy.sum(dim=1)
        return dx, dw, db


def blinear(x: torch.Tensor, w: torch.Tensor,
            b: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Per-client linear: x [C,B,in] @ w[C,in,out] + b[C,out] -> [C,B,out].

    One batched GEMM over the client dimension (hipBLASLt bmm -> MFMA on
    gfx950).  Weights are stored [in, out] so the forward needs no
    transpose; on GPU the custom Function above also keeps the backward
    GEMM operands contiguous (hipBLASLt strided-view fault)."""
    if x.is_cuda:
        return _BLinearFn.apply(x.contiguous(), w.contiguous(), b)
    y = torch.bmm(x, w)
    if b is not None:
        y = y + b.unsqueeze(1)
    return y


def bconv2d(x: torch.Tensor, w: torch.Tensor, clients: int,
            b: Optional[torch.Tensor] = None, stride: int = 1,
            padding: int = 0) -> torch.Tensor:
    """Per-client conv in channel-grouped layout.

    x: [B, C*in_ch, H, W]; w: [C, out_ch, in_ch, kh, kw] -> flattened to
    [C*out_ch, in_ch, kh, kw]; groups=C gives every client its own
    filters in one launch.  Returns [B, C*out_ch, OH, OW].
    """
    c, oc, ic, kh, kw = w.shape
    wf = w.reshape(c * oc, ic, kh, kw)
    bf = b.reshape(c * oc) if b is not None else None
    return F.conv2d(x, wf, bf, stride=stride, padding=padding, groups=clients)


def bgroupnorm(x: torch.Tensor, clients: int, num_groups: int,
               weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    """Per-client GroupNorm in channel-grouped layout.

    x: [B, C*ch, H, W]; weight/bias: [C, ch].  Normalises over each
    client's (ch/num_groups, H, W) group independently, all clients in
    one elementwise pass.
    """
    B, cch, H, W = x.shape
    ch = cch // clients
    xg = x.reshape(B, clients, num_groups, ch // num_groups, H, W)
    mean = xg.mean(dim=(3, 4, 5), keepdim=True)
    var = xg.var(dim=(3, 4, 5), unbiased=False, keepdim=True)
    xn = (xg - mean) * torch.rsqrt(var + eps)
    xn = xn.reshape(B, clients, ch, H, W)
    xn = xn * weight.unsqueeze(0).unsqueeze(-1).unsqueeze(-1) \
        + bias.unsqueeze(0).unsqueeze(-1).unsqueeze(-1)
    return xn.reshape(B, cch, H, W)


def blayernorm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    """Per-client LayerNorm: x [C,B,...,F]; weight/bias [C,F]."""
    if x.is_cuda:
        from ..ops.fused import layernorm
        out = layernorm(x, weight.contiguous(), bias.contiguous(), eps)
        if out is not None:
            return out
    mean = x.mean(dim=-1, keepdim=True)
    var = x.var(dim=-1, unbiased=False, keepdim=True)
    xn = (x - mean) * torch.rsqrt(var + eps)
    extra = x.dim() - 2
    w = weight.reshape(weight.shape[0], *([1] * extra), weight.shape[1])
    b = bias.reshape(bias.shape[0], *([1] * extra), bias.shape[1])
    return xn * w + b
