import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from olearning_sim_amd.ops import load_hip_ops
ops = load_hip_ops(required=True)
C,B,IC,OC,H,st = 250,16,64,64,32,1
x = torch.randn(C,IC,B,H,H,device="cuda",dtype=torch.bfloat16)
w = torch.randn(C,OC,IC,3,3,device="cuda",dtype=torch.bfloat16)*0.05
for _ in range(5):
    y = ops.conv3x3_fwd(x,w,st)
torch.cuda.synchronize()
