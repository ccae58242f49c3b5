import faulthandler, torch
faulthandler.enable()
from olearning_sim_amd.engine import EngineJob, LogicalEngine
job = EngineJob(task_id="b", model_name="bert-base", model_kwargs={},
                clients=125, rounds=1, local_steps=1, batch_size=4,
                lr=0.02, device="cuda:0", dtype="bfloat16",
                vocab_size=30522, seq_len=128, num_classes=0, seed=1)
print("engine init...", flush=True)
eng = LogicalEngine(job)
print("chunk size:", eng._chunk_size(125), flush=True)
for r in range(1):
    rec = eng.run_round(r)
    print("round", r, rec["success"], flush=True)
print("done", flush=True)
