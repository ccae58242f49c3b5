"""Probe: does F.scaled_dot_product_attention handle the strided q/k/v
slices of a packed qkv tensor at large batch (the round-1 workaround
makes 3 contiguous copies per layer per step)?"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch
import torch.nn.functional as F

def run(CB, nh, L, hd, tag):
    g = torch.Generator().manual_seed(0)
    qkv = torch.randn(CB, L, 3, nh, hd, generator=g).to(torch.bfloat16).cuda()
    p = qkv.permute(2, 0, 3, 1, 4)          # [3, CB, nh, L, hd] strided
    q, k, v = p[0], p[1], p[2]
    qc, kc, vc = q.contiguous(), k.contiguous(), v.contiguous()
    a1 = F.scaled_dot_product_attention(q, k, v, is_causal=True)
    a2 = F.scaled_dot_product_attention(qc, kc, vc, is_causal=True)
    torch.cuda.synchronize()
    err = (a1.float() - a2.float()).abs().max().item()
    # grads through the strided path
    qkv.requires_grad_(True)
    p = qkv.permute(2, 0, 3, 1, 4)
    out = F.scaled_dot_product_attention(p[0], p[1], p[2], is_causal=True)
    out.sum().backward()
    torch.cuda.synchronize()
    def t(fn, it=10):
        for _ in range(3): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(it): fn()
        torch.cuda.synchronize(); return (time.perf_counter()-t0)/it*1e3
    ms_view = t(lambda: F.scaled_dot_product_attention(p[0].detach(), p[1].detach(), p[2].detach(), is_causal=True))
    ms_cont = t(lambda: F.scaled_dot_product_attention(
        p[0].detach().contiguous(), p[1].detach().contiguous(),
        p[2].detach().contiguous(), is_causal=True))
    print(f"{tag}: maxerr {err} view {ms_view:.3f} ms vs contig+copy "
          f"{ms_cont:.3f} ms", flush=True)

if __name__ == "__main__":
    run(500, 12, 128, 64, "bert-base C=125 B=4")
    run(64, 12, 128, 64, "small")
    print("SDPA PROBES PASSED", flush=True)
