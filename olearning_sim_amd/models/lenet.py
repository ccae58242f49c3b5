"""LeNet-5 for the CIFAR-10 1k-client config (BASELINE config 2)."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from .base import (ClientBatchedModel, Params, binit, kaiming,
                   blinear, bconv2d)


class LeNet5(ClientBatchedModel):
    name = "lenet"
    num_classes = 10
    act_elems_per_sample = 40_000
    input_shape = (3, 32, 32)

    def __init__(self, in_ch: int = 3, num_classes: int = 10):
        self.in_ch = in_ch
        self.num_classes = num_classes
        self.input_shape = (in_ch, 32, 32)

    def init_global(self, device="cpu", dtype=torch.float32,
                    generator: Optional[torch.Generator] = None) -> Params:
        ic, k = self.in_ch, self.num_classes
        g = generator
        return {
            "conv1.w": kaiming((6, ic, 5, 5), ic * 25, device, dtype, g),
            "conv1.b": binit((6,), ic * 25, device, dtype, g),
            "conv2.w": kaiming((16, 6, 5, 5), 6 * 25, device, dtype, g),
            "conv2.b": binit((16,), 6 * 25, device, dtype, g),
            "fc1.w": binit((400, 120), 400, device, dtype, g),
            "fc1.b": binit((120,), 400, device, dtype, g),
            "fc2.w": binit((120, 84), 120, device, dtype, g),
            "fc2.b": binit((84,), 120, device, dtype, g),
            "fc3.w": binit((84, k), 84, device, dtype, g),
            "fc3.b": binit((k,), 84, device, dtype, g),
        }

    def forward(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        # Fast path: client-channel-first layout + the hand-written
        # valid-conv5x5 MFMA kernels with fused bias+ReLU
        # (ops/csrc/client_conv5.hip) — replaces MIOpen's naive_conv_*
        # fallbacks at thousands of groups (~50% of round GPU time in
        # profiles/fedprox_churn_r01.md).  OLSIM_CONV=miopen forces the
        # grouped reference path.
        import os
        if x.is_cuda and x.dtype == torch.bfloat16 and \
                os.environ.get("OLSIM_CONV", "custom") == "custom":
            from ..ops.fused import hip_ops_available
            if hip_ops_available():
                return self.forward_cbf(params, x)
        # x: [C, B, 3, 32, 32] -> channel-grouped [B, C*3, 32, 32]
        C, B = x.shape[0], x.shape[1]
        xg = x.permute(1, 0, 2, 3, 4).reshape(B, C * self.in_ch, 32, 32)
        h = F.relu(bconv2d(xg, params["conv1.w"], C, params["conv1.b"]))   # 28
        h = F.max_pool2d(h, 2)                                             # 14
        h = F.relu(bconv2d(h, params["conv2.w"], C, params["conv2.b"]))    # 10
        h = F.max_pool2d(h, 2)                                             # 5
        h = h.reshape(B, C, 16 * 5 * 5).permute(1, 0, 2)                   # [C,B,400]
        h = F.relu(blinear(h, params["fc1.w"], params["fc1.b"]))
        h = F.relu(blinear(h, params["fc2.w"], params["fc2.b"]))
        return blinear(h, params["fc3.w"], params["fc3.b"])

    def forward_cbf(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        from ..ops.conv import client_conv5x5, max_pool2x2
        C, B = x.shape[0], x.shape[1]
        h = x.permute(0, 2, 1, 3, 4).contiguous()          # [C, 3, B, 32, 32]
        h = client_conv5x5(h, params["conv1.w"], params["conv1.b"],
                           relu=True)                      # [C, 6, B, 28, 28]
        h = max_pool2x2(h)                                 # [C, 6, B, 14, 14]
        h = client_conv5x5(h, params["conv2.w"], params["conv2.b"],
                           relu=True)                      # [C, 16, B, 10, 10]
        h = max_pool2x2(h)                                 # [C, 16, B, 5, 5]
        h = h.view(C, 16, B, 25).permute(0, 2, 1, 3).reshape(C, B, 400)
        h = F.relu(blinear(h, params["fc1.w"], params["fc1.b"]))
        h = F.relu(blinear(h, params["fc2.w"], params["fc2.b"]))
        return blinear(h, params["fc3.w"], params["fc3.b"])
