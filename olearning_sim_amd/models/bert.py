"""Client-batched BERT for federated next-word prediction
(BASELINE config 5: federated BERT-base, 1k clients, bf16 MFMA GEMMs).

Every client owns a full transformer; weights carry a leading client
dim [C, ...] and all dense math runs as client-batched GEMMs
(torch.bmm -> hipBLASLt MFMA kernels on gfx950).  Attention folds the
client dim into the batch dim of scaled_dot_product_attention.  The LM
head predicts the next token (causal mask), CE via the fused HIP kernel.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from .base import (ClientBatchedModel, Params, binit, blinear, blayernorm)


class _TiedHeadCE(torch.autograd.Function):
    """Fused tied-LM-head + cross-entropy over vocab slices.

    loss = mean_n CE(hs @ tok^T + bias, labels) computed slice-by-slice
    with online softmax statistics — the [C·B·L, V] logits tensor
    (3.9 GB at C=125 for BERT-base) is never materialised, forward or
    backward (docs/ROUND2_DESIGN.md §5).  All GEMMs are client-batched
    bmm on natural-layout operands (the hipBLASLt strided-transpose
    fault, models/base.py, is avoided by materialising per-slice
    transposes of the SMALL dlogits tile only)."""

    SLICE_V = 8192

    @staticmethod
    def forward(ctx, hs, tok, bias, labels):
        # hs [C, N, H] (bf16), tok [C, V, H], bias [C, V], labels [C, N]
        C, N, H = hs.shape
        V = tok.shape[1]
        sv = _TiedHeadCE.SLICE_V
        # hipBLASLt faults on SLICED+TRANSPOSED bmm operands (batch
        # stride > matrix footprint + transpose; tools/bmmprobe.py
        # probe_sliced reproduces it on this stack — full-tensor
        # transposed views are fine).  Materialise tok^T once; its
        # non-transposed slices are safe bmm operands.
        from ..ops.fused import fast_transpose
        tok_t = fast_transpose(tok)                     # [C, H, V]
        m = torch.full((C, N), float("-inf"), device=hs.device,
                       dtype=torch.float32)
        l = torch.zeros(C, N, device=hs.device, dtype=torch.float32)
        zy = torch.zeros(C, N, device=hs.device, dtype=torch.float32)
        lab = labels.view(C, N)
        for s0 in range(0, V, sv):
            s1 = min(s0 + sv, V)
            logits = torch.bmm(hs, tok_t[:, :, s0:s1]).float() \
                + bias[:, s0:s1].float().unsqueeze(1)
            m_new = torch.maximum(m, logits.max(dim=2).values)
            l = l * torch.exp(m - m_new) + \
                torch.exp(logits - m_new.unsqueeze(2)).sum(dim=2)
            m = m_new
            sel = (lab >= s0) & (lab < s1)
            idx = (lab - s0).clamp(0, s1 - s0 - 1)
            zy = torch.where(sel, logits.gather(2, idx.unsqueeze(2))
                             .squeeze(2), zy)
        loss = (torch.log(l) + m - zy).mean()
        ctx.save_for_backward(hs, tok, tok_t, bias, lab, m, l)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        hs, tok, tok_t, bias, lab, m, l = ctx.saved_tensors
        C, N, H = hs.shape
        V = tok.shape[1]
        sv = _TiedHeadCE.SLICE_V
        g = grad_out / (C * N)
        lse = (m + torch.log(l)).unsqueeze(2)           # [C, N, 1]
        dhs = torch.zeros_like(hs)
        dtok = torch.empty_like(tok)
        dbias = torch.empty_like(bias)
        flat_rows = torch.arange(C * N, device=hs.device)
        for s0 in range(0, V, sv):
            s1 = min(s0 + sv, V)
            logits = torch.bmm(hs, tok_t[:, :, s0:s1]).float() \
                + bias[:, s0:s1].float().unsqueeze(1)
            p = torch.exp(logits - lse)                  # softmax slice
            sel = (lab >= s0) & (lab < s1)
            idx = (lab - s0).clamp(0, s1 - s0 - 1)
            pf = p.view(C * N, s1 - s0)
            self_ = sel.view(-1)
            pf[flat_rows[self_], idx.view(-1)[self_]] -= 1.0
            dlogits = (p * g).to(hs.dtype)               # [C, N, s]
            dbias[:, s0:s1] = dlogits.sum(dim=1).to(bias.dtype)
            # tok slice is non-transposed (safe); dlogits' transpose is
            # of a fresh contiguous tensor (full-tensor view, safe)
            dhs += torch.bmm(dlogits, tok[:, s0:s1].to(hs.dtype))
            dtok[:, s0:s1] = torch.bmm(dlogits.transpose(1, 2),
                                       hs).to(tok.dtype)
        return dhs, dtok, dbias, None


def tied_head_ce(hs: torch.Tensor, tok: torch.Tensor, bias: torch.Tensor,
                 labels: torch.Tensor) -> torch.Tensor:
    return _TiedHeadCE.apply(hs, tok, bias, labels)


class BertLM(ClientBatchedModel):
    name = "bert"
    sequence_model = True

    def __init__(self, vocab_size: int = 30522, hidden: int = 768,
                 layers: int = 12, heads: int = 12, seq_len: int = 128,
                 intermediate: Optional[int] = None):
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.layers = layers
        self.heads = heads
        self.seq_len = seq_len
        self.intermediate = intermediate or hidden * 4
        self.num_classes = vocab_size
        self.input_shape = (seq_len,)
        self.act_elems_per_sample = seq_len * hidden * layers * 10

    # ------------------------------------------------------------------
    def init_global(self, device="cpu", dtype=torch.float32,
                    generator: Optional[torch.Generator] = None) -> Params:
        g = generator
        h, inter, v = self.hidden, self.intermediate, self.vocab_size
        p: Params = {}

        def lin(name, out_f, in_f):
            # [in, out] layout: blinear needs no transpose (see base.py)
            p[f"{name}.w"] = binit((in_f, out_f), in_f, device, dtype, g)
            p[f"{name}.b"] = binit((out_f,), in_f, device, dtype, g)

        def ln(name):
            p[f"{name}.g"] = torch.ones(h, device=device, dtype=dtype)
            p[f"{name}.b"] = torch.zeros(h, device=device, dtype=dtype)

        emb = torch.empty(v, h, device=device, dtype=torch.float32)
        emb.normal_(0, 0.02, generator=g)
        p["emb.tok"] = emb.to(dtype)
        pos = torch.empty(self.seq_len, h, device=device, dtype=torch.float32)
        pos.normal_(0, 0.02, generator=g)
        p["emb.pos"] = pos.to(dtype)
        ln("emb.ln")
        for i in range(self.layers):
            pre = f"l{i}"
            lin(f"{pre}.qkv", 3 * h, h)
            lin(f"{pre}.attn_out", h, h)
            ln(f"{pre}.ln1")
            lin(f"{pre}.ffn_in", inter, h)
            lin(f"{pre}.ffn_out", h, inter)
            ln(f"{pre}.ln2")
        # LM head ties to the token embedding (BERT-style decoder bias)
        p["head.bias"] = torch.zeros(v, device=device, dtype=dtype)
        return p

    # ------------------------------------------------------------------
    def encode(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        """Transformer body -> hidden states [C, B*L, H]."""
        # x: [C, B, L] token ids
        C, B, L = x.shape
        h, nh = self.hidden, self.heads
        hd = h // nh
        tok = params["emb.tok"]                    # [C, V, H]
        flat_ids = (x + torch.arange(C, device=x.device)
                    .view(C, 1, 1) * self.vocab_size).reshape(-1)
        emb = tok.reshape(C * self.vocab_size, h)[flat_ids].view(C, B, L, h)
        emb = emb + params["emb.pos"][:, :L].unsqueeze(1)
        hs = blayernorm(emb, params["emb.ln.g"], params["emb.ln.b"])

        for i in range(self.layers):
            pre = f"l{i}"
            qkv = blinear(hs.view(C, B * L, h), params[f"{pre}.qkv.w"],
                          params[f"{pre}.qkv.b"])          # [C, B*L, 3H]
            qkv = qkv.view(C * B, L, 3, nh, hd).permute(2, 0, 3, 1, 4)
            # strided q/k/v slices: verified correct and ~20% faster
            # than materialised copies on this stack (tools/
            # sdpaprobe.py, maxerr 0.0 incl. backward at C*B=500)
            q, k, v = qkv[0], qkv[1], qkv[2]               # [C*B, nh, L, hd]
            att = F.scaled_dot_product_attention(q, k, v, is_causal=True)
            att = att.transpose(1, 2).reshape(C, B * L, h)
            hs = hs + blinear(att, params[f"{pre}.attn_out.w"],
                              params[f"{pre}.attn_out.b"]).view(C, B, L, h)
            hs = blayernorm(hs, params[f"{pre}.ln1.g"], params[f"{pre}.ln1.b"])
            ff = F.gelu(blinear(hs.view(C, B * L, h),
                                params[f"{pre}.ffn_in.w"],
                                params[f"{pre}.ffn_in.b"]))
            hs = hs + blinear(ff, params[f"{pre}.ffn_out.w"],
                              params[f"{pre}.ffn_out.b"]).view(C, B, L, h)
            hs = blayernorm(hs, params[f"{pre}.ln2.g"], params[f"{pre}.ln2.b"])
        return hs.view(C, B * L, h).contiguous()

    def forward(self, params: Params, x: torch.Tensor) -> torch.Tensor:
        C, B, L = x.shape
        hs = self.encode(params, x)
        # tied LM head: logits = hs @ emb^T + bias  -> [C, B, L, V].
        # The transpose is materialised and the GEMM goes through
        # blinear so fwd AND bwd operands are contiguous (hipBLASLt
        # strided-view fault, models/base.py _BLinearFn).
        from ..ops.fused import fast_transpose
        tok_t = fast_transpose(params["emb.tok"])
        logits = blinear(hs, tok_t, params["head.bias"])
        return logits.view(C, B, L, self.vocab_size)

    def loss(self, params: Params, x: torch.Tensor,
             y: torch.Tensor) -> torch.Tensor:
        # y: [C, B, L] next-token targets
        C, B, L = x.shape
        if x.is_cuda:
            # fused tied-head + CE over vocab slices: the [C·B·L, V]
            # logits never materialise (fwd or bwd)
            hs = self.encode(params, x)
            bias = params["head.bias"]
            if bias.dim() == 1:
                bias = bias.unsqueeze(0).expand(C, -1)
            return tied_head_ce(hs, params["emb.tok"], bias,
                                y.reshape(C, B * L))
        logits = self.forward(params, x)
        V = logits.shape[-1]
        from ..ops import cross_entropy_fwd_bwd
        return cross_entropy_fwd_bwd(logits.reshape(C * B * L, V),
                                     y.reshape(-1))


class BertBase(BertLM):
    name = "bert-base"

    def __init__(self, seq_len: int = 128, vocab_size: int = 30522):
        super().__init__(vocab_size=vocab_size, hidden=768, layers=12,
                         heads=12, seq_len=seq_len)


class BertTiny(BertLM):
    """4-layer/128-hidden variant for tests and CPU runs."""
    name = "bert"

    def __init__(self, seq_len: int = 32, vocab_size: int = 1000,
                 hidden: int = 128, layers: int = 2, heads: int = 4):
        super().__init__(vocab_size=vocab_size, hidden=hidden, layers=layers,
                         heads=heads, seq_len=seq_len)
