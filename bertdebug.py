import faulthandler, sys, torch
faulthandler.enable()
from olearning_sim_amd.models import build_model
from olearning_sim_amd.engine.client_manager import FlatParams, replicate_params
from olearning_sim_amd.ops import load_hip_ops
load_hip_ops(required=True)

C, B, L = 2, 2, 128
m = build_model("bert-base", seq_len=L)
gen = torch.Generator().manual_seed(0)
print("init global...", flush=True)
gp = {k: v.cuda() for k, v in m.init_global(generator=gen).items()}
master = FlatParams(gp)
print("replicate...", flush=True)
params = replicate_params(master.cast(torch.bfloat16), C)
x = torch.randint(0, 30522, (C, B, L), device="cuda")
y = torch.randint(0, 30522, (C, B, L), device="cuda")
print("forward...", flush=True)
logits = m.forward(params, x)
torch.cuda.synchronize(); print("forward ok", logits.shape, flush=True)
print("loss...", flush=True)
loss = m.loss(params, x, y)
torch.cuda.synchronize(); print("loss ok", float(loss), flush=True)
print("backward...", flush=True)
g = torch.autograd.grad(loss, list(params.values()), allow_unused=True)
torch.cuda.synchronize(); print("backward ok", flush=True)
