"""Distributed engine worker.

Runs one LogicalEngine rank under ``torch.distributed.run`` — the
execution fabric the NodeClusterManager launches (one process per GPU,
RCCL over xGMI; gloo on CPU).  The reference's analogue is the Ray job
entrypoint ``python3 run_task.py --task '<json>'``
(taskMgr/task_runner.py:69-75); here the payload is the EngineJob as
JSON:

    python -m torch.distributed.run --standalone --local-addr 127.0.0.1 \\
        --nproc-per-node 8 -m olearning_sim_amd.engine.worker \\
        --job-json job.json --result-json out.json

Clients in the job spec are the TOTAL population; each rank takes its
shard (parallel/sharding.py).  Rank 0 writes the aggregate result JSON.
"""

from __future__ import annotations

import argparse
import json
import sys


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--job-json", required=True,
                    help="path to an EngineJob JSON, or an inline JSON string")
    ap.add_argument("--result-json", default="",
                    help="rank 0 writes the run summary here")
    args = ap.parse_args(argv)

    raw = args.job_json
    if raw.strip().startswith("{"):
        spec = json.loads(raw)
    else:
        with open(raw) as f:
            spec = json.load(f)

    from .job import EngineJob
    from .round_loop import LogicalEngine
    from ..parallel import dist as pdist
    from ..parallel.sharding import shard_clients

    ctx = pdist.init_distributed()
    total_clients = int(spec.get("clients", 1))
    lo, hi = shard_clients(total_clients, ctx.rank, max(1, ctx.world_size))
    spec = dict(spec)
    spec["clients"] = max(1, hi - lo)
    # tier/data segments in the spec describe the TOTAL population; after
    # sharding they no longer sum to this rank's clients, so fall back to
    # single-segment accounting rather than miscount
    dropped = False
    for key in ("tier_counts", "data_segments"):
        seg = spec.get(key)
        if seg and sum(int(s[-1]) for s in seg) != spec["clients"]:
            spec.pop(key, None)
            dropped = True
    if dropped:
        spec.pop("dynamic_nums", None)
    known = {f.name for f in EngineJob.__dataclass_fields__.values()}
    job = EngineJob(**{k: v for k, v in spec.items() if k in known})
    job.device = ctx.device

    # rank 0 collects the shaped per-round result rows (the
    # logical_round/logical_result rows the in-process runner writes to
    # the task table; the group monitor replays them on completion)
    round_rows = []
    sink = round_rows.append if ctx.rank == 0 else None
    eng = LogicalEngine(job, dist_ctx=ctx if ctx.enabled else None,
                        result_sink=sink)
    out = eng.run()
    out["total_clients"] = total_clients
    if ctx.rank == 0:
        out["round_rows"] = round_rows
        line = json.dumps({k: v for k, v in out.items()
                           if k not in ("records", "round_rows")})
        print(line)
        if args.result_json:
            with open(args.result_json, "w") as f:
                json.dump(out, f)
    if ctx.enabled:
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
