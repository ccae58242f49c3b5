"""Find who issues the big aten::copy_ calls in a round.

Runs one resnet round under a TorchDispatchMode that logs a python
traceback for every copy_ whose destination exceeds --min-mb.
"""

import argparse
import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.utils._python_dispatch import TorchDispatchMode


class CopySpy(TorchDispatchMode):
    def __init__(self, min_bytes):
        self.min_bytes = min_bytes
        self.seen = {}

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        kwargs = kwargs or {}
        name = str(func)
        if "copy_" in name or "contiguous" in name or "clone" in name:
            t = args[0]
            if isinstance(t, torch.Tensor) and \
                    t.numel() * t.element_size() >= self.min_bytes:
                stack = "".join(traceback.format_stack()[-8:-1])
                key = (name, tuple(t.shape), stack)
                self.seen[key] = self.seen.get(key, 0) + 1
        return func(*args, **kwargs)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--min-mb", type=float, default=200.0)
    ap.add_argument("--clients", type=int, default=1250)
    args = ap.parse_args()

    from olearning_sim_amd.engine.job import EngineJob
    from olearning_sim_amd.engine.round_loop import LogicalEngine
    job = EngineJob(task_id="copyspy", clients=args.clients, rounds=2,
                    model_name="resnet18",
                    model_kwargs={"num_classes": 100}, num_classes=100,
                    local_steps=2, batch_size=16, lr=0.05,
                    dtype="bfloat16", device="cuda:0",
                    dirichlet_alpha=0.1, dynamic_num=10 ** 9)
    eng = LogicalEngine(job)
    eng.run_round(0)                      # warmup / shape specialisation
    torch.cuda.synchronize()
    spy = CopySpy(int(args.min_mb * 1e6))
    with spy:
        eng.run_round(1)
    torch.cuda.synchronize()
    rows = sorted(spy.seen.items(),
                  key=lambda kv: -kv[1])
    for (name, shape, stack), count in rows[:20]:
        print(f"\n=== {name} shape={list(shape)} x{count}")
        print(stack)


if __name__ == "__main__":
    main()
