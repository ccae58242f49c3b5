"""Synthetic non-IID federated data.

There is no network in this environment, so data is synthetic by
construction (BASELINE.json: "synthetic non-IID data / random-init
weights").  Each virtual client owns a label distribution drawn once
from Dirichlet(alpha) over the classes (alpha=0.1 reproduces the
heavily skewed shards of BASELINE config 3); features are sampled fresh
each step on the training device so the data pipeline never leaves the GPU.

Replaces the reference's per-actor zip-download data plumbing
(utils_run_task.py:174-325) — the *shape* of the contract (each client
has its own shard; shards differ in class mix) is preserved, the
transport is gone.
"""

from __future__ import annotations

from typing import Tuple

import torch


class SyntheticFederatedData:
    def __init__(self, clients: int, num_classes: int, input_shape,
                 dirichlet_alpha: float = 0.1, shard_size: int = 64,
                 seed: int = 0, device: str = "cpu",
                 vocab_size: int = 0, seq_len: int = 0):
        self.clients = clients
        self.num_classes = num_classes
        self.input_shape = tuple(input_shape)
        self.shard_size = shard_size
        self.device = torch.device(device)
        self.vocab_size = vocab_size
        self.seq_len = seq_len
        cpu_gen = torch.Generator().manual_seed(seed)
        if vocab_size == 0:
            # per-client class mixture ~ Dirichlet(alpha): sample gammas
            conc = torch.full((clients, num_classes), float(dirichlet_alpha))
            gam = torch._standard_gamma(conc, generator=cpu_gen)
            probs = gam / gam.sum(dim=1, keepdim=True).clamp_min(1e-12)
            # fixed per-client shard labels
            self.labels = torch.multinomial(
                probs, shard_size, replacement=True, generator=cpu_gen
            ).to(torch.int64).to(self.device)
        else:
            self.labels = None
        self._gen = None  # lazily created device generator
        self._seed = seed

    def _step_gen(self, round_idx: int, step: int) -> torch.Generator:
        """Generator seeded per (job seed, round, step): the feature draw
        is independent of how the cohort is chunked, so chunked and
        unchunked training produce bitwise-identical rounds."""
        g = torch.Generator(device=self.device)
        g.manual_seed((self._seed * 1000003 + round_idx * 1009 + step) & 0x7FFFFFFF)
        return g

    def batch(self, client_ids: torch.Tensor, round_idx: int, step: int,
              batch_size: int, dtype: torch.dtype
              ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Return (x [C,B,*input_shape], y [C,B]) for a chunk of clients."""
        C = int(client_ids.numel())
        g = self._step_gen(round_idx, step)
        if self.vocab_size:
            # language modelling: a shared random stream per step, shifted
            # per client so shards differ but stay chunk-invariant
            tok = torch.randint(0, self.vocab_size,
                                (batch_size, self.seq_len + 1),
                                device=self.device, generator=g)
            tok = (tok.unsqueeze(0) + client_ids.to(self.device)
                   .view(C, 1, 1) * 2654435761) % self.vocab_size
            return tok[:, :, :-1], tok[:, :, 1:]
        # classification: fixed per-client shard labels, shared features
        # plus a weak class-conditional shift (training reduces loss)
        idx = ((round_idx * 131 + step * 17
                + torch.arange(batch_size, device=self.device))
               % self.shard_size)
        y = self.labels[client_ids.to(self.device)][:, idx]        # [C, B]
        x = torch.empty((batch_size,) + self.input_shape,
                        device=self.device, dtype=torch.float32)
        x.normal_(generator=g)
        n = x[0].numel()
        if (x.is_cuda and dtype == torch.bfloat16 and n % 8 == 0):
            from ..ops.fused import load_hip_ops
            ops = load_hip_ops()
            if ops is not None:
                # fused broadcast + class shift + bf16 cast
                # (replicate.hip k_synth_batch): the composed form is a
                # stride-0 fp32 broadcast-add plus a second full-size
                # cast copy (~40 ms/round on the flagship)
                out = ops.synth_batch(x.view(batch_size, n),
                                      y.reshape(-1).contiguous(),
                                      0.1 / max(1, self.num_classes), -0.05)
                return out.view(C, batch_size, *self.input_shape), y
        x = x.unsqueeze(0) + 0.1 * (
            y.float().reshape(C, batch_size, *([1] * len(self.input_shape)))
            / max(1, self.num_classes) - 0.5)
        return x.to(dtype), y
