import faulthandler, torch
faulthandler.enable()
from olearning_sim_amd.ops import load_hip_ops, fused
ops = load_hip_ops(required=True)

# 1) CE kernel alone at engine-scale N and K
for N in (512, 4096, 65536):
    logits = torch.randn(N, 30522, device="cuda", dtype=torch.bfloat16)
    labels = torch.randint(0, 30522, (N,), device="cuda")
    lg = logits.requires_grad_(True)
    loss = fused.cross_entropy_fwd_bwd(lg, labels)
    loss.backward()
    torch.cuda.synchronize()
    print("CE ok N=", N, float(loss.detach()), flush=True)

# 2) engine LM data through the loss
from olearning_sim_amd.engine.data import SyntheticFederatedData
d = SyntheticFederatedData(clients=8, num_classes=0, input_shape=(128,),
                           seed=1, device="cuda", vocab_size=30522, seq_len=128)
ids = torch.arange(8)
x, y = d.batch(ids, 0, 0, 4, torch.bfloat16)
print("x", x.shape, x.dtype, "y", y.shape, y.dtype,
      int(y.min()), int(y.max()), flush=True)
logits = torch.randn(8*4*128, 30522, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
loss = fused.cross_entropy_fwd_bwd(logits, y.reshape(-1))
loss.backward()
torch.cuda.synchronize()
print("engine-data CE ok", float(loss.detach()), flush=True)
