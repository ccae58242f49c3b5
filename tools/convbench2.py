import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
from olearning_sim_amd.ops import load_hip_ops
ops = load_hip_ops(required=True)
def t(fn, it=10, wu=3):
    for _ in range(wu): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/it*1000
B=16
for C in (8, 32, 125, 250):
    IC=OC=64; H=32; st=1
    x = torch.randn(C,IC,B,H,H,device="cuda",dtype=torch.bfloat16)
    w = torch.randn(C,OC,IC,3,3,device="cuda",dtype=torch.bfloat16)*0.05
    f = t(lambda: ops.conv3x3_fwd(x,w,st))
    fl = 2*C*B*OC*IC*9*H*H
    print(f"C={C:4d} ic64 h32: fwd {f:7.3f} ms  {fl/f/1e9:6.0f} TF")
