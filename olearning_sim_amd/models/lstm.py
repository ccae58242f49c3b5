"""Client-batched LSTM language model (next-token prediction).

The classic cross-device FL workload (LEAF Shakespeare-style next-char
LSTM) the reference's users run as operator scripts — here it is a
first-class client-batched model: every weight carries a leading client
dimension [C, ...], one `blinear` bmm (hipBLASLt -> MFMA on gfx950)
processes all co-resident clients at once, and the input projection of
the whole sequence is hoisted into a single large GEMM so the
per-timestep recurrence only does the [C,B,H]x[C,H,4H] hidden GEMM.
"""

from __future__ import annotations

from typing import Optional

import torch

from .base import ClientBatchedModel, Params, binit, blinear


class CharLSTM(ClientBatchedModel):
    name = "lstm"
    sequence_model = True

    def __init__(self, vocab_size: int = 90, embed: int = 8,
                 hidden: int = 256, layers: int = 2, seq_len: int = 80):
        self.vocab_size = vocab_size
        self.embed = embed
        self.hidden = hidden
        self.layers = layers
        self.seq_len = seq_len
        self.num_classes = vocab_size
        self.input_shape = (seq_len,)
        self.act_elems_per_sample = seq_len * hidden * (layers * 8 + 2)

    # ------------------------------------------------------------------
    def init_global(self, device="cpu", dtype=torch.float32,
                    generator: Optional[torch.Generator] = None) -> Params:
        g = generator
        v, e, h = self.vocab_size, self.embed, self.hidden
        p: Params = {"emb": binit((v, e), e, device, dtype, g)}
        for i in range(self.layers):
            d_in = e if i == 0 else h
            p[f"l{i}.w_ih"] = binit((d_in, 4 * h), d_in, device, dtype, g)
            p[f"l{i}.w_hh"] = binit((h, 4 * h), h, device, dtype, g)
            p[f"l{i}.b"] = torch.zeros(4 * h, device=device, dtype=dtype)
        p["head.w"] = binit((h, v), h, device, dtype, g)
        p["head.b"] = torch.zeros(v, device=device, dtype=dtype)
        return p

    def param_shapes(self):
        v, e, h = self.vocab_size, self.embed, self.hidden
        shapes = {"emb": (v, e), "head.w": (h, v), "head.b": (v,)}
        for i in range(self.layers):
            d_in = e if i == 0 else h
            shapes[f"l{i}.w_ih"] = (d_in, 4 * h)
            shapes[f"l{i}.w_hh"] = (h, 4 * h)
            shapes[f"l{i}.b"] = (4 * h,)
        return shapes

    # ------------------------------------------------------------------
    def forward(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        # x: [C, B, L] token ids -> logits [C, B, L, V]
        C, B, L = x.shape
        h = self.hidden
        emb = params["emb"]                               # [C, V, E]
        flat_ids = (x + torch.arange(C, device=x.device)
                    .view(C, 1, 1) * self.vocab_size).reshape(-1)
        hs = emb.reshape(C * self.vocab_size, self.embed)[flat_ids] \
            .view(C, B, L, self.embed)

        for i in range(self.layers):
            w_ih, w_hh = params[f"l{i}.w_ih"], params[f"l{i}.w_hh"]
            bias = params[f"l{i}.b"]
            # whole-sequence input projection: one big batched GEMM
            xp = blinear(hs.reshape(C, B * L, hs.shape[-1]), w_ih, bias) \
                .view(C, B, L, 4 * h)
            ht = torch.zeros(C, B, h, device=x.device, dtype=hs.dtype)
            ct = torch.zeros_like(ht)
            outs = []
            for t in range(L):
                gates = xp[:, :, t] + blinear(ht, w_hh)   # [C, B, 4H]
                i_g, f_g, g_g, o_g = gates.split(h, dim=-1)
                i_g = torch.sigmoid(i_g)
                f_g = torch.sigmoid(f_g)
                g_g = torch.tanh(g_g)
                o_g = torch.sigmoid(o_g)
                ct = f_g * ct + i_g * g_g
                ht = o_g * torch.tanh(ct)
                outs.append(ht)
            hs = torch.stack(outs, dim=2)                 # [C, B, L, H]

        logits = blinear(hs.reshape(C, B * L, h), params["head.w"],
                         params["head.b"])
        return logits.view(C, B, L, self.vocab_size)

    def loss(self, params: Params, x: torch.Tensor,
             y: torch.Tensor) -> torch.Tensor:
        # y: [C, B, L] next-token targets
        logits = self.forward(params, x)
        n = logits.shape[0] * logits.shape[1] * logits.shape[2]
        from ..ops import cross_entropy_fwd_bwd
        return cross_entropy_fwd_bwd(
            logits.reshape(n, self.vocab_size), y.reshape(n))
