"""Python surface of the fused HIP ops.

Replaces the per-device subprocess training loop of the reference
(ols_core/taskMgr/utils/utils_run_task.py:481-514) with client-batched
device kernels.  Three bandwidth-bound fused ops carry the optimizer/
aggregation tail of every local-train step:

- fused_sgd_update:      w_c -= lr * (g_c + mu * (w_c - w_global))
                         (mu=0 -> FedAvg local SGD; mu>0 -> FedProx)
- weighted_delta_accum:  delta += sum_c alpha_c * (w_c - w_global)
- apply_aggregate:       w_global += delta / total_weight  (fp32 master)
- cross_entropy_fwd_bwd: fused softmax CE loss + dlogits in one pass

On ROCm these run as hand-written gfx950 kernels (csrc/*.hip, bf16x8 per
lane); on CPU they fall back to eager torch so the control-plane tests
run here.  A CUDA tensor with no extension raises (no silent fallback).
"""

from __future__ import annotations

import os
from typing import Optional, Sequence

import torch

_HIP_OPS = None
_HIP_OPS_TRIED = False

_SO_NAME = "_hip_ops.so"


def _so_path() -> str:
    return os.path.join(os.path.dirname(os.path.abspath(__file__)), _SO_NAME)


def load_hip_ops(required: bool = False):
    """Load the in-tree extension if built. Caches the result."""
    global _HIP_OPS, _HIP_OPS_TRIED
    if _HIP_OPS is not None:
        return _HIP_OPS
    if _HIP_OPS_TRIED and not required:
        return None
    _HIP_OPS_TRIED = True
    path = _so_path()
    if os.path.exists(path):
        torch.ops.load_library(path)
        _HIP_OPS = torch.ops.olsim_hip
        return _HIP_OPS
    if required:
        raise RuntimeError(
            f"olearning_sim_amd HIP extension missing: {path} not built. "
            f"Run `python -m olearning_sim_amd.ops.build` (hipcc, gfx950).")
    return None


def hip_ops_available() -> bool:
    return load_hip_ops() is not None


_OFFS_CACHE = {}
_EMPTY_OFFS = None


def _cached_offsets(n: int, device) -> torch.Tensor:
    """[0, n] block table, cached — avoids a tiny H2D copy per
    parameter per step in the list-form fused ops."""
    key = (n, str(device))
    t = _OFFS_CACHE.get(key)
    if t is None:
        t = torch.tensor([0, n], dtype=torch.int64, device=device)
        _OFFS_CACHE[key] = t
    return t


def _gpu_ops(t: torch.Tensor):
    """Return the HIP op namespace for a GPU tensor; raise if unavailable."""
    if t.is_cuda:
        return load_hip_ops(required=True)
    return None


# ---------------------------------------------------------------------------
def fused_sgd_update(params: Sequence[torch.Tensor],
                     grads: Sequence[torch.Tensor],
                     lr: float,
                     mu: float = 0.0,
                     global_params: Optional[Sequence[torch.Tensor]] = None) -> None:
    """In-place SGD / FedProx step over per-client parameter tensors.

    params[i]: [C, ...] client replicas; grads[i] same shape;
    global_params[i]: [...] (no client dim) — required when mu > 0.
    """
    if not params:
        return
    if params[0].is_cuda:
        for i, (w, g) in enumerate(zip(params, grads)):
            n = w[0].numel() if w.dim() > 1 else w.numel()
            gl = (global_params[i].reshape(-1) if (global_params is not None
                                                  and mu != 0.0)
                  else torch.empty(0, dtype=w.dtype, device=w.device))
            offs = (_cached_offsets(n, w.device) if mu != 0.0
                    else _cached_offsets(0, w.device)[:0])
            fused_sgd_update_flat(w.reshape(-1), g.reshape(-1), gl,
                                  w.shape[0], lr, mu, offs)
        return
    with torch.no_grad():
        for i, (w, g) in enumerate(zip(params, grads)):
            upd = g
            if mu != 0.0 and global_params is not None:
                upd = g + mu * (w - global_params[i].to(w.dtype))
            w.add_(upd, alpha=-lr)


def weighted_delta_accum(delta: Sequence[torch.Tensor],
                         client_params: Sequence[torch.Tensor],
                         global_params: Sequence[torch.Tensor],
                         weights: torch.Tensor,
                         wsum: Optional[float] = None) -> None:
    """delta[i] (fp32, shape [...]) += sum_c weights[c] * (client_params[i][c] - global[i])."""
    if not delta:
        return
    if client_params[0].is_cuda:
        if wsum is None:
            wsum = float(weights.sum())
        for dl, cw, gw in zip(delta, client_params, global_params):
            offs = _cached_offsets(gw.numel(), cw.device)
            weighted_delta_accum_flat(dl.reshape(-1), cw.reshape(-1),
                                      gw.reshape(-1), weights, cw.shape[0],
                                      offs, wsum)
        return
    with torch.no_grad():
        for dl, cw, gw in zip(delta, client_params, global_params):
            diff = cw.float() - gw.float().unsqueeze(0)
            shape = [-1] + [1] * (cw.dim() - 1)
            dl.add_((diff * weights.view(shape)).sum(dim=0))


def apply_aggregate(global_params: Sequence[torch.Tensor],
                    delta: Sequence[torch.Tensor],
                    total_weight: float) -> None:
    """w_global += delta / total_weight, in place (fp32 master weights)."""
    if total_weight == 0:
        return
    with torch.no_grad():
        inv = 1.0 / float(total_weight)
        torch._foreach_add_([g for g in global_params],
                            [d for d in delta], alpha=inv)


# -- flat (single-buffer) forms used by the engine hot path ----------------
#
# Replica layout (engine/client_manager.replicate_flat): the 1-D buffer of
# length C*P is the concatenation over parameter blocks of [C, n_b]; block
# b covers global-master elements offsets[b]:offsets[b+1].

def fused_sgd_update_flat(buf: torch.Tensor, grad: torch.Tensor,
                          global_flat: torch.Tensor, clients: int,
                          lr: float, mu: float,
                          offsets: Optional[torch.Tensor] = None) -> None:
    """buf -= lr * (grad + mu * (buf - replicated(global_flat))), in place."""
    if buf.is_cuda:
        ops = load_hip_ops(required=True)
        ops.fused_sgd_update_flat(
            buf, grad,
            global_flat if mu != 0.0 else torch.empty(0, dtype=buf.dtype, device=buf.device),
            offsets if (mu != 0.0 and offsets is not None) else
            torch.empty(0, dtype=torch.int64, device=buf.device),
            int(clients), float(lr), float(mu))
        return
    with torch.no_grad():
        if mu == 0.0 or offsets is None:
            buf.add_(grad, alpha=-lr)
            return
        offs = offsets.tolist()
        for b in range(len(offs) - 1):
            g0, g1 = offs[b], offs[b + 1]
            n = g1 - g0
            blk = buf[clients * g0:clients * g1].view(clients, n)
            gblk = grad[clients * g0:clients * g1].view(clients, n)
            prox = blk - global_flat[g0:g1].unsqueeze(0)
            blk.add_(gblk + mu * prox, alpha=-lr)


def weighted_delta_accum_flat(delta: torch.Tensor, buf: torch.Tensor,
                              global_flat: torch.Tensor,
                              weights: torch.Tensor, clients: int,
                              offsets: Optional[torch.Tensor] = None,
                              wsum: Optional[float] = None) -> None:
    """delta[j] += sum_c weights[c] * (buf[c,j] - global_flat[j]) per block."""
    if buf.is_cuda:
        ops = load_hip_ops(required=True)
        if wsum is None:
            wsum = float(weights.sum())
        ops.weighted_delta_accum_flat(delta, buf, global_flat,
                                      weights.float(), offsets, int(clients),
                                      float(wsum))
        return
    with torch.no_grad():
        offs = offsets.tolist()
        w = weights.float().view(clients, 1)
        for b in range(len(offs) - 1):
            g0, g1 = offs[b], offs[b + 1]
            n = g1 - g0
            blk = buf[clients * g0:clients * g1].view(clients, n).float()
            diff = blk - global_flat[g0:g1].float().unsqueeze(0)
            delta[g0:g1].add_((diff * w).sum(dim=0))


class _CrossEntropyFn(torch.autograd.Function):
    """Fused CE over [N, K] logits (HIP kernel fwd computes loss AND
    dlogits in one pass; backward is a scale)."""

    @staticmethod
    def forward(ctx, logits: torch.Tensor, labels: torch.Tensor):
        ops = load_hip_ops(required=True)
        row_loss, dlogits = ops.cross_entropy_fwd_bwd(
            logits.contiguous(), labels)
        ctx.save_for_backward(dlogits)
        return row_loss.mean()

    @staticmethod
    def backward(ctx, grad_out):
        (dlogits,) = ctx.saved_tensors
        return dlogits * grad_out, None


def cross_entropy_fwd_bwd(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over all rows. logits [N, K] (bf16/f32), labels [N] int64."""
    if logits.is_cuda:
        return _CrossEntropyFn.apply(logits, labels)
    return torch.nn.functional.cross_entropy(logits.float(), labels)


class _GroupNormActFn(torch.autograd.Function):
    """Fused per-client GroupNorm (+residual +ReLU) on channel-grouped
    activations [B, C*ch, H, W] (ops/csrc/groupnorm.hip)."""

    @staticmethod
    def forward(ctx, x, res, gamma, beta, clients, groups, eps, relu):
        ops = load_hip_ops(required=True)
        has_res = res is not None
        res_in = res.contiguous() if has_res else torch.empty(0, dtype=x.dtype,
                                                              device=x.device)
        y, mean, rstd = ops.groupnorm_fwd(x.contiguous(), res_in, gamma,
                                          beta, clients, groups, eps, relu)
        ctx.save_for_backward(x, y, mean, rstd, gamma)
        ctx.meta = (clients, groups, has_res, relu)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, mean, rstd, gamma = ctx.saved_tensors
        clients, groups, has_res, relu = ctx.meta
        ops = load_hip_ops(required=True)
        dx, dres, dgamma, dbeta = ops.groupnorm_bwd(
            x, y, dy, mean, rstd, gamma, clients, groups, has_res, relu)
        return (dx, dres.view_as(x) if has_res else None,
                dgamma.to(gamma.dtype), dbeta.to(gamma.dtype),
                None, None, None, None)


def groupnorm_act(x: torch.Tensor, clients: int, groups: int,
                  gamma: torch.Tensor, beta: torch.Tensor,
                  res: Optional[torch.Tensor] = None, relu: bool = False,
                  eps: float = 1e-5) -> torch.Tensor:
    """y = relu?(gn(x; clients, groups)*gamma + beta [+ res]).

    x 4-D [B, C*ch, H, W] (channel-grouped) or 5-D [C, ch, B, H, W]
    (client-channel-first, the conv-kernel layout).  GPU: one fused HIP
    kernel each way.  CPU: composed torch ops."""
    if x.is_cuda:
        return _GroupNormActFn.apply(x, res, gamma.contiguous(),
                                     beta.contiguous(), clients, groups,
                                     eps, relu)
    if x.dim() == 5:
        C, ch, B, H, W = x.shape
        cg = ch // groups
        xg = x.view(C, groups, cg, B, H, W)
        mean = xg.mean(dim=(2, 4, 5), keepdim=True)
        var = xg.var(dim=(2, 4, 5), unbiased=False, keepdim=True)
        y = ((xg - mean) * torch.rsqrt(var + eps)).view(C, ch, B, H, W)
        y = y * gamma.view(C, ch, 1, 1, 1) + beta.view(C, ch, 1, 1, 1)
    else:
        from ..models.base import bgroupnorm
        y = bgroupnorm(x, clients, groups, gamma, beta, eps)
    if res is not None:
        y = y + res
    return torch.nn.functional.relu(y) if relu else y


def fast_transpose(x: torch.Tensor) -> torch.Tensor:
    """[B, M, N] -> contiguous [B, N, M] via the LDS-tiled transpose
    kernel (torch's transpose+contiguous runs the generic strided copy
    at a fraction of the roofline on large batched matrices)."""
    if x.is_cuda and x.dim() == 3 and \
            x.dtype in (torch.bfloat16, torch.float32):
        ops = load_hip_ops()
        if ops is not None:
            return ops.transpose2d(x.contiguous())
    return x.transpose(1, 2).contiguous()


class _LayerNormFn(torch.autograd.Function):
    """Fused per-client LayerNorm over the trailing dim
    (ops/csrc/layernorm.hip; one wave per row, fp32 dgamma/dbeta
    partials through workgroup LDS)."""

    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        ops = load_hip_ops(required=True)
        y, mean, rstd = ops.layernorm_fwd(x, gamma, beta, eps)
        ctx.save_for_backward(x, gamma, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd = ctx.saved_tensors
        ops = load_hip_ops(required=True)
        dx, dgamma, dbeta = ops.layernorm_bwd(x, dy.contiguous(), gamma,
                                              mean, rstd)
        return dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype), None


def layernorm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
              eps: float = 1e-5):
    """Per-client LayerNorm: x [C, ..., H], gamma/beta [C, H].
    Fused kernel on GPU when H%8==0 and rows-per-client %4==0;
    None -> caller should use the composed fallback."""
    shape = x.shape
    C, H = shape[0], shape[-1]
    n = x.numel() // (C * H)
    if not (x.is_cuda and H % 8 == 0 and n % 4 == 0 and H <= 1536
            and x.dtype in (torch.bfloat16, torch.float32)
            and hip_ops_available()):
        return None
    y = _LayerNormFn.apply(x.reshape(C, n, H).contiguous(), gamma, beta, eps)
    return y.view(shape)
