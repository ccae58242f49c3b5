import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
"""Kernel-level conv benchmark: custom MFMA vs MIOpen grouped."""
import time, torch, torch.nn.functional as F
from olearning_sim_amd.ops import load_hip_ops
ops = load_hip_ops(required=True)

def t(fn, it=10, wu=3):
    for _ in range(wu): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/it*1000

C, B = 250, 16
SH = [(64,64,32,1),(128,128,16,1),(256,256,8,1),(512,512,4,1),(64,128,32,2)]
print("shape                 fwd_v2    dgrad    wgrad  | miopen_f  miopen_d  miopen_w")
for IC,OC,H,st in SH:
    x = torch.randn(C,IC,B,H,H,device="cuda",dtype=torch.bfloat16)
    w = torch.randn(C,OC,IC,3,3,device="cuda",dtype=torch.bfloat16)*0.05
    OHW = (H//st)
    dy = torch.randn(C,OC,B,OHW,OHW,device="cuda",dtype=torch.bfloat16)
    f  = t(lambda: ops.conv3x3_fwd(x,w,st))
    d  = t(lambda: ops.conv3x3_dgrad(dy,w,H,H,st))
    wg = t(lambda: ops.conv3x3_wgrad(x,dy,st))
    # MIOpen reference
    xg = x.permute(2,0,1,3,4).reshape(B,C*IC,H,H).contiguous().requires_grad_(True)
    wf = w.reshape(C*OC,IC,3,3).detach().requires_grad_(True)
    y0 = F.conv2d(xg,wf,stride=st,padding=1,groups=C)
    g0 = torch.randn_like(y0)
    mf = t(lambda: F.conv2d(xg,wf,stride=st,padding=1,groups=C))
    def dg():
        y = F.conv2d(xg,wf,stride=st,padding=1,groups=C)
        torch.autograd.grad(y,[xg],g0)
    def wgr():
        y = F.conv2d(xg,wf,stride=st,padding=1,groups=C)
        torch.autograd.grad(y,[wf],g0)
    md = t(dg,5,2)-mf; mw = t(wgr,5,2)-mf
    fl = 2*C*B*OC*IC*9*OHW*OHW
    print(f"ic{IC:3d} oc{OC:3d} h{H:2d} s{st}: {f:7.2f} ({fl/f/1e9:5.0f}TF) {d:7.2f} {wg:7.2f} | {mf:8.2f} {md:8.2f} {mw:8.2f}")
