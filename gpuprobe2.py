"""Probe grouped-conv execution paths: layout x dtype x find-mode."""
import os, sys, time, torch, torch.nn.functional as F

dev = "cuda:0"

def bench(fn, iters=4, warmup=3):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

SHAPES = [(64, 64, 32), (128, 128, 16), (256, 256, 8), (512, 512, 4)]
C, B = 125, 16

def run(tag, dtype, channels_last):
    tot_f = tot_b = 0.0
    for ic, oc, hw in SHAPES:
        x = torch.randn(B, C*ic, hw, hw, device=dev, dtype=dtype)
        w = torch.randn(C*oc, ic, 3, 3, device=dev, dtype=dtype)
        if channels_last:
            x = x.to(memory_format=torch.channels_last)
            w = w.to(memory_format=torch.channels_last)
        x.requires_grad_(True); w.requires_grad_(True)
        y = F.conv2d(x, w, groups=C, padding=1)
        g = torch.randn_like(y)
        f = bench(lambda: F.conv2d(x, w, groups=C, padding=1))
        def bwd():
            y = F.conv2d(x, w, groups=C, padding=1)
            torch.autograd.grad(y, [x, w], g)
        bw = bench(bwd) - f
        flops = 2 * B * C * oc * ic * 9 * hw * hw
        tot_f += f; tot_b += bw
        print(f"  {tag} ic{ic} oc{oc} hw{hw}: fwd {f:7.2f} ms ({flops/f/1e9:6.1f} TF) fwd+bwd extra {bw:8.2f} ms")
    print(f"  {tag} TOTAL fwd {tot_f:.1f} ms, bwd-extra {tot_b:.1f} ms")

print("== nchw bf16 (default) =="); run("nchw-bf16", torch.bfloat16, False)
print("== nhwc bf16 =="); run("nhwc-bf16", torch.bfloat16, True)
print("== nhwc fp16 =="); run("nhwc-fp16", torch.float16, True)
