import sys, time, torch, torch.nn.functional as F
dev="cuda:0"; B,C,ic,oc,hw = 16,250,512,512,4
mode = sys.argv[1] if len(sys.argv)>1 else "clean"
if mode == "pressure":
    hog = torch.empty(40*2**30, dtype=torch.uint8, device=dev)  # 40 GB
x = torch.randn(B, C*ic, hw, hw, device=dev, dtype=torch.bfloat16, requires_grad=True)
w = torch.randn(C*oc, ic, 3, 3, device=dev, dtype=torch.bfloat16, requires_grad=True)
y = F.conv2d(x, w, groups=C, padding=1); g = torch.randn_like(y)
def dgrad():
    y = F.conv2d(x, w, groups=C, padding=1)
    torch.autograd.grad(y, [x], g)
dgrad(); dgrad(); torch.cuda.synchronize()
t0=time.perf_counter()
for _ in range(3): dgrad()
torch.cuda.synchronize()
print(f"[{mode}] dgrad steady: {(time.perf_counter()-t0)/3*1000:.2f} ms")
