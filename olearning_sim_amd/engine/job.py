"""Engine job specification.

The runnable form of a task's logical simulation: what the reference
serialises into the Ray job entrypoint JSON (taskMgr/task_runner.py:69-75
submitting run_task.py --task '<json>').  Built either directly (bench,
tests) or from a TaskConfig by task/runner.py.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List


@dataclass
class EngineJob:
    task_id: str = "job"
    model_name: str = "mlp"
    model_kwargs: Dict[str, Any] = field(default_factory=dict)

    # population & per-round work
    clients: int = 10               # population size (this rank's shard)
    cohort_size: int = 0            # clients trained per round (0 = all)
    rounds: int = 1
    local_steps: int = 2            # E: local SGD steps per client per round
    batch_size: int = 8             # per-client local batch
    lr: float = 0.05
    prox_mu: float = 0.0            # FedProx proximal coefficient (0 = FedAvg)

    # execution
    dtype: str = "float32"          # compute dtype of client replicas
    device: str = "cpu"
    chunk_clients: int = 0          # cohort chunk co-resident on the GPU (0 = auto)
    seed: int = 1234

    # data
    num_classes: int = 10
    dirichlet_alpha: float = 0.1
    shard_size: int = 64
    vocab_size: int = 0             # >0 switches to LM data
    seq_len: int = 0

    # behaviour simulation (deviceflow): arrival/offline/drop shaping
    behavior_strategy: str = ""     # gradient-house strategy JSON ("" = none)

    # ordered operator list executed each round (reference
    # operatorflow.operators, run_task.py:228): (name, kind) with kind
    # in {"train", "evaluate", "checkpoint"}
    operators: List[Any] = field(default_factory=lambda: [("train", "train")])

    # operator-flow round gates (reference flow_setting.start/stop)
    flow_start_strategy: str = ""
    flow_stop_strategy: str = ""
    flow_wait_interval: float = 1.0
    flow_total_timeout: float = 0.0
    flow_work_dir: str = ""

    # global-model evaluation (the reference's post-train operators):
    # every N rounds run the aggregated model on a held-out synthetic
    # batch and report loss/accuracy (0 = off)
    eval_every: int = 0
    eval_batch: int = 64

    # bookkeeping / checkpointing
    checkpoint_dir: str = ""
    model_update_style: str = ""    # e.g. "{task_id}_{current_round}_result_model.safetensors"
    save_every_round: bool = False
    # resume from the newest saved round artifact (crash recovery of an
    # interrupted run).  OFF by default: a fresh submission of the same
    # task id must start from round 0 like the reference's run_task.
    resume: bool = False

    # failure-tolerance accounting (reference total_simulation semantics)
    data_name: str = "data_0"
    device_tier: str = "high"
    dynamic_num: int = 0            # tolerated failures per round (total)
    # per-tier populations for this rank: [(tier, clients), ...] —
    # client ids are assigned to tiers in order (prefix ranges); empty
    # means a single job.device_tier tier of all clients
    tier_counts: List[Any] = field(default_factory=list)
    dynamic_nums: List[int] = field(default_factory=list)  # per segment
    # full (data x tier) segment list for multi-data tasks:
    # [(data_name, tier, clients), ...] in client-id prefix order;
    # overrides tier_counts when non-empty
    data_segments: List[Any] = field(default_factory=list)

    def resolved_cohort(self) -> int:
        return self.cohort_size if self.cohort_size > 0 else self.clients

    def torch_dtype(self):
        import torch
        return {"float32": torch.float32, "bfloat16": torch.bfloat16,
                "float16": torch.float16}[self.dtype]
