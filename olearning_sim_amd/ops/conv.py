"""Client-batched convolution ops in client-channel-first layout.

Layout [C, ch, B, H, W]: per (client, channel) the [B, H, W] block is
contiguous — the implicit-GEMM-friendly layout of
ops/csrc/client_conv.hip (custom MFMA kernels for 3x3), and 1x1 convs
collapse to one batched GEMM over clients (torch.bmm -> hipBLASLt).

CPU fallback: grouped F.conv2d after a layout permute (reference
semantics, used by tests and the CPU plumbing config).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from .fused import load_hip_ops


def _cpu_conv3x3(x: torch.Tensor, w: torch.Tensor, stride: int) -> torch.Tensor:
    # x [C, IC, B, H, W], w [C, OC, IC, 3, 3] -> y [C, OC, B, OH, OW]
    C, IC, B, H, W = x.shape
    OC = w.shape[1]
    xg = x.permute(2, 0, 1, 3, 4).reshape(B, C * IC, H, W)
    wf = w.reshape(C * OC, IC, 3, 3)
    y = F.conv2d(xg, wf, stride=stride, padding=1, groups=C)
    OH, OW = y.shape[-2:]
    return y.reshape(B, C, OC, OH, OW).permute(1, 2, 0, 3, 4).contiguous()


class _Conv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, stride):
        ops = load_hip_ops(required=True)
        y = ops.conv3x3_fwd(x, w, stride)
        ctx.save_for_backward(x, w)
        ctx.stride = stride
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        ops = load_hip_ops(required=True)
        dy = dy.contiguous()
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = ops.conv3x3_dgrad(dy, w, x.shape[3], x.shape[4], ctx.stride)
        if ctx.needs_input_grad[1]:
            dw = ops.conv3x3_wgrad(x, dy, ctx.stride)
        return dx, dw, None


def _pad(x: torch.Tensor, pad: int) -> torch.Tensor:
    """Zero-pad the trailing two dims (single-pass kernel; torch's
    constant_pad_nd is a fill + strided-copy double pass)."""
    import os
    if (x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 2 == 0
            and os.environ.get("OLSIM_PAD", "") != "torch"):
        ops = load_hip_ops()
        if ops is not None:
            return ops.pad2d(x.contiguous(), pad)
    return F.pad(x, (pad, pad, pad, pad))


class _Conv3x3PadFn(torch.autograd.Function):
    """v6 padded path: the kernels gather from a 1-element zero halo so
    the implicit-im2col reads carry no bounds masks (client_conv2.hip).
    The padded activation is saved for backward (wgrad reuses it)."""

    @staticmethod
    def forward(ctx, x, w, stride):
        ops = load_hip_ops(required=True)
        x_pad = _pad(x, 1)
        y = ops.conv3x3_fwd_p(x_pad, w, stride)
        ctx.save_for_backward(x_pad, w)
        ctx.stride = stride
        return y

    @staticmethod
    def backward(ctx, dy):
        x_pad, w = ctx.saved_tensors
        ops = load_hip_ops(required=True)
        dy = dy.contiguous()
        dx = dw = None
        H, W = x_pad.shape[3] - 2, x_pad.shape[4] - 2
        if ctx.needs_input_grad[0]:
            dy_pad = _pad(dy, 1)
            dx = ops.conv3x3_dgrad_p(dy_pad, w, H, W, ctx.stride)
        if ctx.needs_input_grad[1]:
            dw = ops.conv3x3_wgrad_p(x_pad, dy, ctx.stride)
        return dx, dw, None


def _v6_ok(ops, x: torch.Tensor, w: torch.Tensor, stride: int) -> bool:
    import os
    if os.environ.get("OLSIM_CONV_V", "") == "5":
        return False
    C, IC, B, H, W = x.shape
    return bool(ops.conv3x3_v6_ok(IC, w.shape[1], B, H, W, stride))


def client_conv3x3(x: torch.Tensor, w: torch.Tensor,
                   stride: int = 1) -> torch.Tensor:
    """y[C,OC,B,OH,OW] = conv3x3(x[C,IC,B,H,W], w[C,OC,IC,3,3]), pad 1."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        ops = load_hip_ops(required=True)
        if _v6_ok(ops, x, w, stride):
            return _Conv3x3PadFn.apply(x.contiguous(), w.contiguous(), stride)
        return _Conv3x3Fn.apply(x.contiguous(), w.contiguous(), stride)
    return _cpu_conv3x3(x, w, stride)


# ---------------------------------------------------------------------------
# 5x5 VALID convolution (LeNet family, client_conv5.hip)

_NTAB_CACHE: dict = {}


def _ntab(B: int, OH: int, OW: int, H: int, W: int,
          device: torch.device) -> torch.Tensor:
    """int32 [B*OH*OW] plane offsets b*H*W + oh*W + ow (gather base per
    output position; the planes are not powers of two, so the kernels
    index a table instead of dividing per element)."""
    key = (B, OH, OW, H, W, str(device))
    t = _NTAB_CACHE.get(key)
    if t is None:
        n = torch.arange(B * OH * OW, dtype=torch.int64)
        b, q = n // (OH * OW), n % (OH * OW)
        t = (b * H * W + (q // OW) * W + (q % OW)).to(torch.int32).to(device)
        if len(_NTAB_CACHE) < 64:
            _NTAB_CACHE[key] = t
    return t


def _cpu_conv5x5(x, w, b, relu):
    # x [C, IC, B, H, W], w [C, OC, IC, 5, 5] valid conv
    C, IC, B, H, W = x.shape
    OC = w.shape[1]
    xg = x.permute(2, 0, 1, 3, 4).reshape(B, C * IC, H, W)
    y = F.conv2d(xg, w.reshape(C * OC, IC, 5, 5), b.reshape(C * OC),
                 groups=C)
    OH, OW = y.shape[-2:]
    y = y.reshape(B, C, OC, OH, OW).permute(1, 2, 0, 3, 4).contiguous()
    return F.relu(y) if relu else y


class _Conv5x5Fn(torch.autograd.Function):
    """Valid 5x5 conv + bias (+ReLU) on the MFMA kernels.  ReLU backward
    uses the saved post-activation output (y > 0 mask)."""

    @staticmethod
    def forward(ctx, x, w, b, relu):
        ops = load_hip_ops(required=True)
        C, IC, B, H, W = x.shape
        nt = _ntab(B, H - 4, W - 4, H, W, x.device)
        y = ops.conv5x5_fwd(x, w, b, nt, relu)
        ctx.save_for_backward(x, w, y)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        ops = load_hip_ops(required=True)
        C, IC, B, H, W = x.shape
        dy = dy.contiguous()
        if ctx.relu:
            if dy.numel() % 8 == 0:
                dy = ops.relu_mask(dy, y)     # one pass (replicate.hip)
            else:
                dy = dy * (y > 0).to(dy.dtype)
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dy_pad = _pad(dy, 4)
            nt = _ntab(B, H, W, H + 4, W + 4, x.device)
            dx = ops.conv5x5_dgrad(dy_pad, w, nt, H, W)
        if ctx.needs_input_grad[1]:
            nt = _ntab(B, H - 4, W - 4, H, W, x.device)
            dw = ops.conv5x5_wgrad(x, dy, nt)
        db = dy.sum(dim=(2, 3, 4)) if ctx.needs_input_grad[2] else None
        return dx, dw, db, None


def conv5x5_supported(x: torch.Tensor, w: torch.Tensor) -> bool:
    if not (x.is_cuda and x.dtype == torch.bfloat16):
        return False
    C, IC, B, H, W = x.shape
    if w.shape[1] > 16 or IC > 16:
        return False
    return (B * (H - 4) * (W - 4)) % 32 == 0


def client_conv5x5(x: torch.Tensor, w: torch.Tensor, b: torch.Tensor,
                   relu: bool = False) -> torch.Tensor:
    """y[C,OC,B,H-4,W-4] = valid conv5x5(x[C,IC,B,H,W], w[C,OC,IC,5,5])
    + bias[C,OC], optional fused ReLU."""
    if conv5x5_supported(x, w):
        return _Conv5x5Fn.apply(x.contiguous(), w.contiguous(),
                                b.contiguous(), relu)
    return _cpu_conv5x5(x, w, b, relu)


class _Pool2x2Fn(torch.autograd.Function):
    """2x2/2 max pool over the trailing two dims (pool2x2.hip): the
    forward saves a 2-bit argmax per output, the backward writes every
    input position once (no atomics, no zeroing pass) — torch's generic
    nchw pool kernels measured ~8x/15x off roofline on LeNet shapes."""

    @staticmethod
    def forward(ctx, x):
        ops = load_hip_ops(required=True)
        y, arg = ops.pool2x2_fwd(x)
        ctx.save_for_backward(arg)
        return y

    @staticmethod
    def backward(ctx, dy):
        (arg,) = ctx.saved_tensors
        ops = load_hip_ops(required=True)
        return ops.pool2x2_bwd(dy.contiguous(), arg)


def max_pool2x2(x: torch.Tensor) -> torch.Tensor:
    """Max-pool 2x2 stride 2 over the last two dims (H, W even)."""
    if x.is_cuda and x.dtype in (torch.bfloat16, torch.float32):
        return _Pool2x2Fn.apply(x.contiguous())
    shape = x.shape
    flat = x.reshape(-1, 1, shape[-2], shape[-1])
    out = F.max_pool2d(flat, 2)
    return out.reshape(*shape[:-2], shape[-2] // 2, shape[-1] // 2)


class _Subsample2Fn(torch.autograd.Function):
    """Every-2nd-pixel subsample of the trailing dims (pool2x2.hip):
    torch's slice backward is a full zero-fill plus a strided scatter
    through the generic 5-D kernels; here bwd writes the input once."""

    @staticmethod
    def forward(ctx, x):
        ops = load_hip_ops(required=True)
        ctx.hw = x.shape[-2:]
        return ops.subsample2(x.contiguous())

    @staticmethod
    def backward(ctx, dy):
        ops = load_hip_ops(required=True)
        H, W = ctx.hw
        return ops.subsample2_bwd(dy.contiguous(), H, W)


def client_conv1x1(x: torch.Tensor, w: torch.Tensor,
                   stride: int = 1) -> torch.Tensor:
    """1x1 conv = one batched GEMM: y[C,OC,n] = w[C,OC,IC] @ x[C,IC,n]."""
    C, IC, B, H, W = x.shape
    OC = w.shape[1]
    if stride != 1:
        if (stride == 2 and x.is_cuda and H % 2 == 0 and W % 4 == 0
                and x.dtype in (torch.bfloat16, torch.float32)):
            x = _Subsample2Fn.apply(x)
        else:
            x = x[:, :, :, ::stride, ::stride].contiguous()
        H, W = x.shape[-2:]
    y = torch.bmm(w.reshape(C, OC, IC), x.reshape(C, IC, B * H * W))
    return y.view(C, OC, B, H, W)
