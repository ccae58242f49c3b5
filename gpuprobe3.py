"""Probe the ResNet-18 bench's exact conv shapes for naive-kernel fallbacks."""
import time, torch, torch.nn.functional as F
dev="cuda:0"; C,B = 125,16

def bench(fn, iters=3, warmup=2):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1000

# (ic, oc, hw_in, k, stride)
SHAPES = [
    (3,   64, 32, 3, 1, "stem"),
    (64,  64, 32, 3, 1, "s0"),
    (64, 128, 32, 3, 2, "s1.c1"),
    (64, 128, 32, 1, 2, "s1.down"),
    (128,128, 16, 3, 1, "s1.c2"),
    (128,256, 16, 3, 2, "s2.c1"),
    (128,256, 16, 1, 2, "s2.down"),
    (256,256,  8, 3, 1, "s2.c2"),
    (256,512,  8, 3, 2, "s3.c1"),
    (256,512,  8, 1, 2, "s3.down"),
    (512,512,  4, 3, 1, "s3.c2"),
]
for ic,oc,hw,k,s,tag in SHAPES:
    pad = 1 if k==3 else 0
    x = torch.randn(B, C*ic, hw, hw, device=dev, dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(C*oc, ic, k, k, device=dev, dtype=torch.bfloat16, requires_grad=True)
    y = F.conv2d(x, w, groups=C, stride=s, padding=pad); g = torch.randn_like(y)
    f = bench(lambda: F.conv2d(x, w, groups=C, stride=s, padding=pad))
    def dgrad():
        y = F.conv2d(x, w, groups=C, stride=s, padding=pad)
        torch.autograd.grad(y, [x], g, retain_graph=False)
    def wgrad():
        y = F.conv2d(x, w, groups=C, stride=s, padding=pad)
        torch.autograd.grad(y, [w], g, retain_graph=False)
    d = bench(dgrad)-f; wg = bench(wgrad)-f
    print(f"{tag:8s} ic{ic:3d}->oc{oc:3d} hw{hw:2d} k{k} s{s}: fwd {f:8.2f}  dgrad {d:8.2f}  wgrad {wg:8.2f} ms")
