import time, torch, torch.nn.functional as F
dev="cuda:0"; B=16
def bench(fn, iters=3, warmup=2):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1000
for C in (125, 160, 200, 250, 320):
    for ic,oc,hw in ((64,64,32),(512,512,4)):
        x = torch.randn(B, C*ic, hw, hw, device=dev, dtype=torch.bfloat16, requires_grad=True)
        w = torch.randn(C*oc, ic, 3, 3, device=dev, dtype=torch.bfloat16, requires_grad=True)
        y = F.conv2d(x, w, groups=C, padding=1); g = torch.randn_like(y)
        f = bench(lambda: F.conv2d(x, w, groups=C, padding=1))
        def dgrad():
            y = F.conv2d(x, w, groups=C, padding=1)
            torch.autograd.grad(y, [x], g)
        d = bench(dgrad)-f
        print(f"C={C:4d} ic{ic:3d} hw{hw:2d}: fwd {f:8.2f} dgrad {d:8.2f} ms")
