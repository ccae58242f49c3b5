"""Quick perf probes on the GPU box (diagnostics, not the bench)."""
import time, torch, torch.nn.functional as F

def t(fn, iters=5, warmup=2):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

dev = "cuda:0"
print(torch.cuda.get_device_name(0), torch.cuda.get_device_properties(0).total_memory/2**30, "GiB")

# grouped conv probe: C clients, B batch, ch channels
for C in (125, 500, 1250):
    B, ic, oc = 16, 64, 64
    x = torch.randn(B, C*ic, 32, 32, device=dev, dtype=torch.bfloat16)
    w = torch.randn(C*oc, ic, 3, 3, device=dev, dtype=torch.bfloat16)
    ms = t(lambda: F.conv2d(x, w, groups=C, padding=1))
    flops = 2 * B * C * oc * ic * 9 * 32 * 32
    print(f"groupedconv C={C}: {ms:.2f} ms  {flops/ms/1e9:.1f} TF/s")

# bmm probe (client-batched linear)
for C in (1250,):
    B, m, n = 16, 512, 100
    x = torch.randn(C, B, m, device=dev, dtype=torch.bfloat16)
    w = torch.randn(C, n, m, device=dev, dtype=torch.bfloat16)
    ms = t(lambda: torch.bmm(x, w.transpose(1,2)))
    print(f"bmm C={C} {B}x{m}x{n}: {ms:.3f} ms")

# im2col alternative for grouped conv
for C in (1250,):
    B, ic, oc = 16, 64, 64
    x = torch.randn(C, B, ic, 32, 32, device=dev, dtype=torch.bfloat16)
    w = torch.randn(C, oc, ic*9, device=dev, dtype=torch.bfloat16)
    def im2col_conv():
        u = F.unfold(x.reshape(C*B, ic, 32, 32), 3, padding=1)  # [CB, ic*9, 1024]
        u = u.reshape(C, B, ic*9, 1024).permute(0, 2, 1, 3).reshape(C, ic*9, B*1024)
        return torch.bmm(w, u)
    ms = t(im2col_conv)
    flops = 2 * B * C * oc * ic * 9 * 1024
    print(f"im2col+bmm C={C}: {ms:.2f} ms  {flops/ms/1e9:.1f} TF/s")
