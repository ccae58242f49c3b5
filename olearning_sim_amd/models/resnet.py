"""ResNet-18 (CIFAR variant, GroupNorm) — the flagship FedAvg CNN.

BASELINE config 3: FedAvg ResNet-18 on CIFAR-100 non-IID, 10k clients
across 8 GPUs.  BatchNorm is replaced with GroupNorm(8) as is standard
for federated training (per-client running stats are meaningless when
every client re-starts from the global model each round); all clients'
convs run as one grouped conv per layer in the channel-grouped layout
[B, C*ch, H, W] (models/base.py).
"""

from __future__ import annotations

from typing import Optional

import torch

from .base import (ClientBatchedModel, Params, binit, kaiming,
                   blinear, bconv2d)

_STAGES = (64, 128, 256, 512)
_GN_GROUPS = 8


class ResNet18(ClientBatchedModel):
    name = "resnet18"
    num_classes = 100
    act_elems_per_sample = 1_600_000
    input_shape = (3, 32, 32)

    def __init__(self, num_classes: int = 100, in_ch: int = 3,
                 width_mult: float = 1.0):
        self.num_classes = num_classes
        self.in_ch = in_ch
        self.widths = [int(w * width_mult) for w in _STAGES]
        self.input_shape = (in_ch, 32, 32)

    # ------------------------------------------------------------------
    def init_global(self, device="cpu", dtype=torch.float32,
                    generator: Optional[torch.Generator] = None) -> Params:
        g = generator
        p: Params = {}

        def conv(name, oc, ic, k):
            p[f"{name}.w"] = kaiming((oc, ic, k, k), ic * k * k, device, dtype, g)

        def gn(name, ch):
            p[f"{name}.g"] = torch.ones(ch, device=device, dtype=dtype)
            p[f"{name}.b"] = torch.zeros(ch, device=device, dtype=dtype)

        w = self.widths
        conv("stem", w[0], self.in_ch, 3)
        gn("stem.gn", w[0])
        in_c = w[0]
        for s, out_c in enumerate(w):
            for blk in range(2):
                pre = f"s{s}.b{blk}"
                ic = in_c if blk == 0 else out_c
                conv(f"{pre}.c1", out_c, ic, 3)
                gn(f"{pre}.gn1", out_c)
                conv(f"{pre}.c2", out_c, out_c, 3)
                gn(f"{pre}.gn2", out_c)
                if blk == 0 and (ic != out_c or s > 0):
                    conv(f"{pre}.down", out_c, ic, 1)
                    gn(f"{pre}.gndown", out_c)
            in_c = out_c
        p["fc.w"] = binit((w[3], self.num_classes), w[3], device, dtype, g)
        p["fc.b"] = binit((self.num_classes,), w[3], device, dtype, g)
        return p

    # ------------------------------------------------------------------
    def _block(self, params: Params, x: torch.Tensor, C: int, pre: str,
               stride: int, has_down: bool) -> torch.Tensor:
        from ..ops.fused import groupnorm_act
        h = bconv2d(x, params[f"{pre}.c1.w"], C, stride=stride, padding=1)
        h = groupnorm_act(h, C, _GN_GROUPS, params[f"{pre}.gn1.g"],
                          params[f"{pre}.gn1.b"], relu=True)
        h = bconv2d(h, params[f"{pre}.c2.w"], C, stride=1, padding=1)
        if has_down:
            sc = bconv2d(x, params[f"{pre}.down.w"], C, stride=stride)
            sc = groupnorm_act(sc, C, _GN_GROUPS, params[f"{pre}.gndown.g"],
                               params[f"{pre}.gndown.b"])
        else:
            sc = x
        # fused: relu(gn(h) + sc)
        return groupnorm_act(h, C, _GN_GROUPS, params[f"{pre}.gn2.g"],
                             params[f"{pre}.gn2.b"], res=sc, relu=True)

    # -- fast path: client-channel-first layout + hand-written MFMA
    # conv kernels (ops/conv.py / ops/csrc/client_conv.hip) -------------
    def _block_cbf(self, params: Params, x: torch.Tensor, C: int, pre: str,
                   stride: int, has_down: bool) -> torch.Tensor:
        from ..ops.conv import client_conv3x3, client_conv1x1
        from ..ops.fused import groupnorm_act
        h = client_conv3x3(x, params[f"{pre}.c1.w"], stride)
        h = groupnorm_act(h, C, _GN_GROUPS, params[f"{pre}.gn1.g"],
                          params[f"{pre}.gn1.b"], relu=True)
        h = client_conv3x3(h, params[f"{pre}.c2.w"], 1)
        if has_down:
            sc = client_conv1x1(x, params[f"{pre}.down.w"], stride)
            sc = groupnorm_act(sc, C, _GN_GROUPS, params[f"{pre}.gndown.g"],
                               params[f"{pre}.gndown.b"])
        else:
            sc = x
        return groupnorm_act(h, C, _GN_GROUPS, params[f"{pre}.gn2.g"],
                             params[f"{pre}.gn2.b"], res=sc, relu=True)

    def forward_cbf(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        from ..ops.conv import client_conv3x3
        from ..ops.fused import groupnorm_act
        C, B = x.shape[0], x.shape[1]
        h = x.permute(0, 2, 1, 3, 4).contiguous()    # [C, 3, B, 32, 32]
        h = client_conv3x3(h, params["stem.w"], 1)
        h = groupnorm_act(h, C, _GN_GROUPS, params["stem.gn.g"],
                          params["stem.gn.b"], relu=True)
        for s in range(4):
            stride = 1 if s == 0 else 2
            h = self._block_cbf(params, h, C, f"s{s}.b0", stride, s > 0)
            h = self._block_cbf(params, h, C, f"s{s}.b1", 1, False)
        # h: [C, 512, B, 4, 4] -> avg pool -> [C, B, 512]
        h = h.mean(dim=(3, 4)).permute(0, 2, 1)
        return blinear(h, params["fc.w"], params["fc.b"])

    def forward(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        from ..ops.fused import groupnorm_act
        # Default: the hand-written client-batched implicit-GEMM MFMA
        # conv kernels (ops/csrc/client_conv2.hip v6 padded family —
        # mask-free gathers, XCD-aware tiling, parity-decomposed
        # stride-2 dgrad).  Measured vs MIOpen grouped conv at C=250
        # (tools/convbench3.py, profiles/): all-shape conv total
        # 30.9 ms vs 57.1 ms (0.54x).  OLSIM_CONV=miopen forces the
        # grouped-conv reference path.
        import os
        if x.is_cuda and x.dtype == torch.bfloat16 and \
                os.environ.get("OLSIM_CONV", "custom") == "custom":
            from ..ops.fused import hip_ops_available
            if hip_ops_available():
                return self.forward_cbf(params, x)
        # x: [C, B, 3, 32, 32]
        C, B = x.shape[0], x.shape[1]
        h = x.permute(1, 0, 2, 3, 4).reshape(B, C * self.in_ch, 32, 32)
        h = bconv2d(h, params["stem.w"], C, stride=1, padding=1)
        h = groupnorm_act(h, C, _GN_GROUPS, params["stem.gn.g"],
                          params["stem.gn.b"], relu=True)
        for s in range(4):
            stride = 1 if s == 0 else 2
            has_down = s > 0
            h = self._block(params, h, C, f"s{s}.b0", stride, has_down)
            h = self._block(params, h, C, f"s{s}.b1", 1, False)
        # h: [B, C*512, 4, 4] -> global average pool -> [C, B, 512]
        h = h.mean(dim=(2, 3)).reshape(B, C, self.widths[3]).permute(1, 0, 2)
        return blinear(h, params["fc.w"], params["fc.b"])
