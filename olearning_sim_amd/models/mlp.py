"""Two-layer MLP for the MNIST plumbing config (BASELINE config 1)."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from .base import ClientBatchedModel, Params, binit, blinear


class MLP(ClientBatchedModel):
    name = "mlp"
    num_classes = 10
    act_elems_per_sample = 2048
    input_shape = (784,)

    def __init__(self, in_features: int = 784, hidden: int = 200,
                 num_classes: int = 10):
        self.in_features = in_features
        self.hidden = hidden
        self.num_classes = num_classes
        self.input_shape = (in_features,)

    def init_global(self, device="cpu", dtype=torch.float32,
                    generator: Optional[torch.Generator] = None) -> Params:
        d, h, k = self.in_features, self.hidden, self.num_classes
        return {
            "fc1.w": binit((d, h), d, device, dtype, generator),
            "fc1.b": binit((h,), d, device, dtype, generator),
            "fc2.w": binit((h, k), h, device, dtype, generator),
            "fc2.b": binit((k,), h, device, dtype, generator),
        }

    def forward(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        # x: [C, B, in_features]
        h = F.relu(blinear(x, params["fc1.w"], params["fc1.b"]))
        return blinear(h, params["fc2.w"], params["fc2.b"])
