"""Client-batched local training.

One call trains a *chunk* of co-resident clients for E local steps from
the current global model and accumulates their weighted deltas — the
MI355X replacement for the reference's per-phone subprocess loop
(utils_run_task.py:481-514, one `os.system` per virtual device).

Hot path per local step:
  forward/backward through the client-batched model (grouped convs /
  batched MFMA GEMMs), then ONE fused SGD/FedProx update kernel over the
  contiguous [C*P] replica buffer (ops/csrc/fused_update.hip).
After the E steps, ONE weighted-delta reduction kernel folds
(w_c - w_global) * alpha_c into the fp32 delta accumulator
(ops/csrc/aggregate.hip).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from ..models.base import ClientBatchedModel, Params
from ..ops import fused
from .client_manager import FlatParams, replicate_params
from .data import SyntheticFederatedData


class LocalTrainer:
    def __init__(self, model: ClientBatchedModel, master: FlatParams,
                 data: SyntheticFederatedData, lr: float, prox_mu: float,
                 local_steps: int, batch_size: int, dtype: torch.dtype,
                 report_loss: bool = True):
        # report_loss=False avoids a device sync per chunk (bench path)
        self.report_loss = report_loss
        self.model = model
        self.master = master
        self.data = data
        self.lr = lr
        self.prox_mu = prox_mu
        self.local_steps = local_steps
        self.batch_size = batch_size
        self.dtype = dtype
        # cast of the fp32 master into the compute dtype; row-major
        # param-major layout identical to master.flat
        self._cast_flat: Optional[torch.Tensor] = None

    def begin_round(self) -> None:
        """Re-cast the master once per round (clients replicate from it)."""
        self._cast_flat = self.master.flat.to(self.dtype)

    def _cast_params(self) -> Params:
        assert self._cast_flat is not None, "begin_round() not called"
        return self.master.views_of(self._cast_flat)

    def train_chunk(self, client_ids: torch.Tensor, weights: torch.Tensor,
                    round_idx: int, delta_flat: torch.Tensor,
                    wsum: Optional[float] = None) -> Dict[str, float]:
        """Train one chunk of clients; accumulate weighted deltas.

        client_ids: [C] int64; weights: [C] fp32 aggregation weights
        (0 for clients whose gradient the behaviour model drops);
        delta_flat: fp32 [P] accumulator with the master's layout.
        """
        C = int(client_ids.numel())
        cast = self._cast_params()
        params = replicate_params(cast, C)
        plist = list(params.values())
        glist = [cast[k] for k in params]
        last_loss = 0.0
        for step in range(self.local_steps):
            x, y = self.data.batch(client_ids, round_idx, step,
                                   self.batch_size, self.dtype)
            loss = self.model.loss(params, x, y)
            # loss is the mean over C*B rows; each client's SGD step needs
            # the gradient of ITS OWN per-client mean, i.e. d(loss*C)/dw_c
            # — this also makes training invariant to the chunking.
            grads = torch.autograd.grad(loss * C, plist, allow_unused=True)
            with torch.no_grad():
                live = [(p, g) for p, g in zip(plist, grads) if g is not None]
                fused.fused_sgd_update(
                    [p for p, _ in live], [g for _, g in live],
                    lr=self.lr, mu=self.prox_mu,
                    global_params=[cast[k] for k, g in
                                   zip(params, grads) if g is not None])
            if self.report_loss and step == self.local_steps - 1:
                last_loss = float(loss.detach())
            del grads
        with torch.no_grad():
            delta_views = list(self.master.views_of(delta_flat).values())
            fused.weighted_delta_accum(
                delta_views, [p.detach() for p in plist], glist, weights,
                wsum=wsum if wsum is not None else float(weights.sum()))
        return {"loss": last_loss, "clients": C}
