"""The simulation round loop — the hot path.

MI355X-native equivalent of the reference's RayRunner
(ols_core/taskMgr/run_task.py:212-322): for each operator-flow round,
select the cohort, train every cohort client locally (client-batched on
the GPU instead of one subprocess per phone), aggregate the weighted
deltas (RCCL all-reduce across GPUs), update the global model, record
per-round success/failed counts with the reference's tolerance
semantics, and checkpoint under the reference's templated name.

Behaviour simulation (the deviceflow gradient house): an offline mask
removes clients before training (they count as failed machine-times, the
arrival/spike curves), a drop mask zeroes a trained client's aggregation
weight (message dropped in flight — the device itself still succeeded,
matching deviceflow semantics where drops happen after NotifyComplete).
"""

from __future__ import annotations

import time
from typing import Any, Callable, Dict, List, Optional

import torch

from ..models import build_model
from ..ops import fused
from ..parallel import dist as pdist
from .checkpoint import save_checkpoint, load_checkpoint, latest_round
from .client_manager import FlatParams
from .data import SyntheticFederatedData
from .job import EngineJob
from .local_train import LocalTrainer

# signature: sampler(round_idx, cohort_size) -> (offline_mask[C], drop_mask[C]) bool tensors
BehaviorFn = Callable[[int, int], tuple]


class LogicalEngine:
    def __init__(self, job: EngineJob,
                 dist_ctx: Optional[pdist.DistContext] = None,
                 behavior: Optional[BehaviorFn] = None,
                 result_sink: Optional[Callable[[Dict[str, Any]], None]] = None,
                 deviceflow=None, perf=None, script_ops=None):
        # script_ops: {operator_name: ScriptOperator} — user script-file
        # operators executed as sharded subprocesses (engine/script_op.py)
        self.script_ops = script_ops or {}
        # deviceflow: gradient-house service handle — the engine drives
        # the reference's NotifyStart/NotifyComplete lifecycle per round
        # (run_task.py:234-308) and publishes per-chunk summary messages
        # to the inbound room for outbound consumers; perf: a
        # PerformanceManager recording per-round metrics.
        self.deviceflow = deviceflow
        self.perf = perf
        self.job = job
        self.ctx = dist_ctx or pdist.DistContext(device=job.device)
        self.device = torch.device(self.ctx.device if dist_ctx else job.device)
        self.dtype = job.torch_dtype()
        self.result_sink = result_sink
        self.stop_requested = False

        self.model = build_model(job.model_name, **job.model_kwargs)
        gen = torch.Generator().manual_seed(job.seed)
        global_params = self.model.init_global(
            device="cpu", dtype=torch.float32, generator=gen)
        global_params = {k: v.to(self.device) for k, v in global_params.items()}
        self.master = FlatParams(global_params)
        if self.ctx.enabled:
            # all ranks seed identically, but broadcast pins exact equality
            pdist.broadcast_flat(self.master.flat)

        self.data = SyntheticFederatedData(
            clients=job.clients, num_classes=job.num_classes,
            input_shape=self.model.input_shape,
            dirichlet_alpha=job.dirichlet_alpha, shard_size=job.shard_size,
            seed=job.seed + 7919 * self.ctx.rank, device=str(self.device),
            vocab_size=job.vocab_size, seq_len=job.seq_len)
        self.trainer = LocalTrainer(
            self.model, self.master, self.data, lr=job.lr,
            prox_mu=job.prox_mu, local_steps=job.local_steps,
            batch_size=job.batch_size, dtype=self.dtype,
            report_loss=result_sink is not None)
        if behavior is None and job.behavior_strategy:
            from ..deviceflow.sampler import BehaviorSampler
            behavior = BehaviorSampler(job.behavior_strategy,
                                       seed=job.seed + self.ctx.rank,
                                       device=str(self.device))
        self.behavior = behavior

        from .operatorflow import OperatorFlow
        self.flow = OperatorFlow(
            job.task_id,
            start_strategy=job.flow_start_strategy,
            stop_strategy=job.flow_stop_strategy,
            wait_interval=job.flow_wait_interval,
            total_timeout=job.flow_total_timeout,
            work_dir=job.flow_work_dir or job.checkpoint_dir or ".")
        # double-buffered delta accumulator: round r's RCCL all-reduce
        # overlaps the host-side round tail (stats/result rows/flow
        # gates/cohort selection) and is waited only where the master is
        # next READ (_finish_aggregate) — semantics identical, comm cost
        # hidden (SURVEY §5 "overlap the collective")
        self._delta = self.master.zeros_like_flat()
        self._delta_alt: Optional[torch.Tensor] = None
        self._pending_agg = None     # (work, delta_buf, total_weight)
        # totals across rounds (reference logical_result accounting)
        self.success_total = 0
        self.failed_total = 0
        self.last_operator = "train"
        # (data x tier) client-id prefix segments (reference: success/
        # failed vectors per data target and device tier)
        if job.data_segments:
            segs = [(d, t, int(n)) for d, t, n in job.data_segments]
        elif job.tier_counts:
            segs = [(job.data_name, t, int(n)) for t, n in job.tier_counts]
        else:
            segs = [(job.data_name, job.device_tier, job.clients)]
        self.segments = segs
        self.tier_names = [t for _, t, _ in segs]
        bounds = [0]
        for _, _, n in segs:
            bounds.append(bounds[-1] + n)
        self.tier_bounds = bounds          # len = S+1; covers [0, clients)
        self.tier_dynamic = (list(job.dynamic_nums)
                             if job.dynamic_nums else [job.dynamic_num])

        # crash resume: load the newest per-round artifact and continue
        # from the next round (reference model_update_style download)
        self.start_round = 0
        if job.resume and job.checkpoint_dir and job.save_every_round:
            r = latest_round(job.checkpoint_dir, job.task_id,
                             job.model_update_style)
            if r >= 0:
                sd = load_checkpoint(job.checkpoint_dir, job.task_id, r,
                                     job.model_update_style,
                                     device=str(self.device))
                if sd is not None:
                    self.master.load_state_dict(sd)
                    self.start_round = r + 1

    # ------------------------------------------------------------------
    def _chunk_size(self, cohort: int) -> int:
        """Clients per chunk.  Computed ONCE per engine (cached): the
        free-memory probe must not move between rounds or kernel shapes
        churn (shape-specialised kernels re-tune, fixed-shape padding
        re-pads a different tail).  Chunks are equalised so a split
        cohort never pads a tiny tail to a full chunk."""
        cached = getattr(self, "_chunk_cache", None)
        if cached is not None and cached[0] == cohort:
            return cached[1]
        if self.job.chunk_clients > 0:
            chunk = min(self.job.chunk_clients, cohort)
        elif self.device.type != "cuda":
            chunk = min(cohort, 64)
        else:
            # bound replica+grad+activation memory by HBM actually free
            # at engine start (torch's cache may already hold the rest)
            p = self.master.numel()
            bytes_per_client = p * self.dtype.itemsize * 2  # weights + grad
            bytes_per_client += (self.job.batch_size
                                 * self.model.act_elems_per_sample
                                 * self.dtype.itemsize)
            free, total = torch.cuda.mem_get_info(self.device)
            free += torch.cuda.memory_reserved(self.device)
            budget = int(min(free * 0.8, total * 0.5))
            max_chunk = max(1, min(cohort, budget // max(1, bytes_per_client)))
            n_chunks = (cohort + max_chunk - 1) // max_chunk
            chunk = (cohort + n_chunks - 1) // n_chunks
        self._chunk_cache = (cohort, chunk)
        return chunk

    def select_cohort(self, round_idx: int) -> torch.Tensor:
        """Deterministic rotating window over this rank's population
        (the reference enumerates machine-times exhaustively per round;
        a cohort < population rotates so every client participates)."""
        job = self.job
        cohort = min(job.resolved_cohort(), job.clients)
        start = (round_idx * cohort) % job.clients
        ids = (start + torch.arange(cohort)) % job.clients
        return ids.to(torch.int64)

    # ------------------------------------------------------------------
    def evaluate_global(self, round_idx: int) -> Dict[str, float]:
        """Run the aggregated global model on a held-out synthetic batch
        (the reference's evaluate-style operator)."""
        self._finish_aggregate()
        job = self.job
        with torch.no_grad():
            cast = {k: v.to(self.dtype).unsqueeze(0)
                    for k, v in self.master.views.items()}
            eval_ids = torch.zeros(1, dtype=torch.int64)
            x, y = self.data.batch(eval_ids, round_idx, 10_000,
                                   job.eval_batch, self.dtype)
            logits = self.model.forward(cast, x)
            k = logits.shape[-1]
            flat = logits.reshape(-1, k).float()
            labels = y.reshape(-1)
            loss = float(torch.nn.functional.cross_entropy(flat, labels))
            acc = float((flat.argmax(-1) == labels).float().mean())
        return {"eval_loss": loss, "eval_acc": acc}

    def _finish_aggregate(self) -> None:
        """Wait any in-flight delta all-reduce and fold it into the
        master (called wherever the master is next read)."""
        if self._pending_agg is None:
            return
        work, delta, total_weight = self._pending_agg
        self._pending_agg = None
        if work is not None:
            work.wait()
        if total_weight > 0:
            fused.apply_aggregate([self.master.flat], [delta], total_weight)

    def _op_train(self, round_idx: int,
                  op_name: str = "train") -> Dict[str, Any]:
        """The training operator: one cohort pass + aggregation."""
        self._finish_aggregate()
        job = self.job
        ids = self.select_cohort(round_idx)
        cohort = int(ids.numel())
        flow_id = None
        if self.deviceflow is not None and job.behavior_strategy:
            flow_id = self.deviceflow.notify_start(
                job.task_id, op_name, round_idx, "logical_simulation",
                strategy=job.behavior_strategy)

        if self.behavior is not None:
            offline, dropped = self.behavior(round_idx, cohort)
        else:
            offline = torch.zeros(cohort, dtype=torch.bool)
            dropped = torch.zeros(cohort, dtype=torch.bool)
        offline = offline.cpu()
        dropped = dropped.cpu()
        active_ids = ids[~offline]
        active_drop = dropped[~offline]

        self.trainer.begin_round()
        self._delta.zero_()
        weights_all = (~active_drop).float()
        losses: List[float] = []
        trained = 0
        n_active = int(active_ids.numel())
        # chunk size derives from the FIXED cohort, not this round's
        # churn-varying active count, so kernel shapes stay constant
        chunk = self._chunk_size(min(job.resolved_cohort(), job.clients))
        for lo in range(0, n_active, chunk):
            cid = active_ids[lo:lo + chunk]
            w_cpu = weights_all[lo:lo + chunk]
            if cid.numel() < chunk and n_active > 0:
                # pad the tail chunk to the full chunk size with
                # zero-weight repeats: every chunk then has the SAME
                # shape, so shape-specialised kernels (MIOpen Find,
                # captured graphs) run once per job instead of once per
                # round as churn varies the cohort
                pad = chunk - int(cid.numel())
                cid = torch.cat([cid, cid[-1:].repeat(pad)])
                w_cpu = torch.cat([w_cpu, torch.zeros(pad)])
            w = w_cpu.to(self.device)
            stats = self.trainer.train_chunk(cid.to(self.device), w,
                                             round_idx, self._delta,
                                             wsum=float(w_cpu.sum()))
            trained += min(stats["clients"], n_active - lo)
            if stats["loss"]:
                losses.append(stats["loss"])
            if flow_id is not None:
                # per-chunk summary into the gradient house (the per-
                # client tensors aggregate in-engine; the message plane
                # carries chunk summaries for outbound consumers)
                self.deviceflow.publish(flow_id, "logical_simulation",
                                        payload={"round": round_idx,
                                                 "clients": stats["clients"],
                                                 "loss": stats["loss"]})

        local_weight = float(weights_all.sum())
        # per-tier success/failed (client ids map to tiers by prefix range)
        T = len(self.tier_names)
        succ_t = [0] * T
        fail_t = [0] * T
        for t in range(T):
            lo, hi = self.tier_bounds[t], self.tier_bounds[t + 1]
            in_tier = (ids >= lo) & (ids < hi)
            fail_t[t] = int((in_tier & offline).sum())
            succ_t[t] = int((in_tier & ~offline).sum())
        success_local = sum(succ_t)
        failed_local = sum(fail_t)

        # cross-GPU aggregation: one RCCL all-reduce of the flat delta +
        # one small all-reduce carrying (weight, per-tier succ/fail)
        if self.ctx.enabled:
            stats = torch.tensor([local_weight] + succ_t + fail_t,
                                 dtype=torch.float64, device=self.device)
            work = pdist.all_reduce_flat(self._delta, async_op=True)
            pdist.all_reduce_flat(stats)      # small, waited (round status)
            total_weight = float(stats[0])
            succ_t = [int(x) for x in stats[1:1 + T]]
            fail_t = [int(x) for x in stats[1 + T:1 + 2 * T]]
            success = sum(succ_t)
            failed = sum(fail_t)
            # defer the big delta wait+apply: the collective overlaps
            # everything until the master is next read; swap to the
            # alternate delta buffer so the next round's accumulation
            # never races the in-flight reduce
            self._pending_agg = (work, self._delta, total_weight)
            if self._delta_alt is None:
                self._delta_alt = self.master.zeros_like_flat()
            self._delta, self._delta_alt = self._delta_alt, self._delta
        else:
            total_weight = local_weight
            success, failed = success_local, failed_local
            if total_weight > 0:
                fused.apply_aggregate([self.master.flat], [self._delta],
                                      total_weight)

        self.success_total += success
        self.failed_total += failed

        if flow_id is not None:
            self.deviceflow.drain_inbound()
            self.deviceflow.notify_complete(job.task_id, op_name, round_idx,
                                            "logical_simulation")
        dyn = self.tier_dynamic
        if len(dyn) != T:
            dyn = [job.dynamic_num] * T
        # per-segment tolerance uniformly (reference checks each tier's
        # failed > dynamic_nums; job.dynamic_num sums across all
        # data/tiers and would inflate the single-segment tolerance)
        round_failed = any(f > d for f, d in zip(fail_t, dyn))
        return {
            "success": success,
            "failed": failed,
            "success_per_tier": succ_t,
            "failed_per_tier": fail_t,
            "trained": trained,
            "loss": sum(losses) / len(losses) if losses else None,
            "round_failed": round_failed,
        }

    def _op_script(self, round_idx: int, name: str) -> Dict[str, Any]:
        """User script-file operator: sharded subprocess execution of
        the staged entry file (reference Actor loop_run,
        utils_run_task.py:481-514).  Counts feed the same per-tier
        tolerance accounting as the in-process train path."""
        from .script_op import per_tier_counts
        op = self.script_ops.get(name)
        if op is None:
            raise RuntimeError(f"script operator {name!r} not staged")
        flow_id = None
        if self.deviceflow is not None and self.job.behavior_strategy:
            flow_id = self.deviceflow.notify_start(
                self.job.task_id, name, round_idx, "logical_simulation",
                strategy=self.job.behavior_strategy)
        model_path = None
        if round_idx > 0 and self.job.checkpoint_dir \
                and self.job.save_every_round:
            import os
            from .checkpoint import checkpoint_name
            cand = os.path.join(
                self.job.checkpoint_dir,
                checkpoint_name(self.job.task_id, round_idx - 1,
                                self.job.model_update_style))
            if os.path.exists(cand):
                model_path = cand
        res = op.run_round(round_idx, model_path=model_path)
        if flow_id is not None:
            self.deviceflow.publish(flow_id, "logical_simulation",
                                    payload={"round": round_idx,
                                             "success": res["success"],
                                             "failed": res["failed"]})
            self.deviceflow.drain_inbound()
            self.deviceflow.notify_complete(self.job.task_id, name,
                                            round_idx, "logical_simulation")
        succ_t, fail_t = per_tier_counts(res["failed_ranges"],
                                         self.tier_bounds)
        self.success_total += res["success"]
        self.failed_total += res["failed"]
        dyn = self.tier_dynamic
        if len(dyn) != len(succ_t):
            dyn = [self.job.dynamic_num] * len(succ_t)
        round_failed = (res["failed"] > self.job.dynamic_num
                        if len(succ_t) == 1
                        else any(f > d for f, d in zip(fail_t, dyn)))
        return {"success": res["success"], "failed": res["failed"],
                "success_per_tier": succ_t, "failed_per_tier": fail_t,
                "round_failed": round_failed}

    def run_round(self, round_idx: int) -> Dict[str, Any]:
        """Execute the round's ordered operator list
        (reference run_task.py:228-311: per operator -> deviceflow
        NotifyStart, work, NotifyComplete, result accumulation)."""
        job = self.job
        t_round = time.time()
        record: Dict[str, Any] = {"round": round_idx, "success": 0,
                                  "failed": 0, "trained": 0, "loss": None,
                                  "round_failed": False,
                                  "success_per_tier": [0] * len(self.tier_names),
                                  "failed_per_tier": [0] * len(self.tier_names)}
        ops = job.operators or [("train", "train")]
        for entry in ops:
            name, kind = (entry if isinstance(entry, (tuple, list))
                          else (entry, "train"))
            if kind == "train":
                record.update(self._op_train(round_idx, name))
            elif kind == "script":
                record.update(self._op_script(round_idx, name))
            elif kind == "evaluate":
                record.update(self.evaluate_global(round_idx))
            elif kind == "checkpoint":
                self._finish_aggregate()
                if self.ctx.rank == 0 and job.checkpoint_dir:
                    record["checkpoint"] = save_checkpoint(
                        job.checkpoint_dir, job.task_id, round_idx,
                        self.master.state_dict(), job.model_update_style)
            self.last_operator = name
        if job.eval_every > 0 and (round_idx + 1) % job.eval_every == 0                 and "eval_acc" not in record:
            record.update(self.evaluate_global(round_idx))
        if self.perf is not None:
            self.perf.record_round(job.task_id, round_idx,
                                   time.time() - t_round, record["success"],
                                   loss=record["loss"])
            if "eval_acc" in record:
                self.perf.record(job.task_id, "eval_acc",
                                 record["eval_acc"], round_idx)
        if job.save_every_round and job.checkpoint_dir                 and "checkpoint" not in record:
            self._finish_aggregate()
            if self.ctx.rank == 0:
                record["checkpoint"] = save_checkpoint(
                    job.checkpoint_dir, job.task_id, round_idx,
                    self.master.state_dict(), job.model_update_style)
        return record

    # ------------------------------------------------------------------
    def run(self) -> Dict[str, Any]:
        job = self.job
        records: List[Dict[str, Any]] = []
        t0 = time.time()
        for r in range(self.start_round, job.rounds):
            if self.stop_requested:
                break
            self.flow.start(r)         # operator-flow start gate
            rec = self.run_round(r)
            self.flow.stop(r)          # operator-flow stop gate
            records.append(rec)
            if self.result_sink is not None:
                self.result_sink(self._round_result(rec))
            if rec["round_failed"]:
                break
        self._finish_aggregate()
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        elapsed = time.time() - t0
        rounds_done = len(records)
        world = max(1, self.ctx.world_size)
        return {
            "task_id": job.task_id,
            "rounds": rounds_done,
            "elapsed_s": elapsed,
            "rounds_per_s": rounds_done / elapsed if elapsed > 0 else 0.0,
            "clients_per_s": (self.success_total / elapsed) if elapsed > 0 else 0.0,
            "success_total": self.success_total,
            "failed_total": self.failed_total,
            "final_loss": records[-1]["loss"] if records else None,
            "world_size": world,
            "records": records,
        }

    def _result_entries(self, rec: Dict[str, Any]) -> List[Dict[str, Any]]:
        """Group the per-segment counts back into one result entry per
        data target (run_task.py analyze_results:149-210 shape)."""
        succ = rec.get("success_per_tier", [rec["success"]])
        fail = rec.get("failed_per_tier", [rec["failed"]])
        order: List[str] = []
        by_data: Dict[str, Dict[str, list]] = {}
        for i, (dn, tier, _n) in enumerate(self.segments):
            if dn not in by_data:
                by_data[dn] = {"devices": [], "success_num": [],
                               "failed_num": []}
                order.append(dn)
            t = by_data[dn]
            t["devices"].append(tier)
            t["success_num"].append(succ[i] if i < len(succ) else 0)
            t["failed_num"].append(fail[i] if i < len(fail) else 0)
        return [{"name": dn, "simulation_target": by_data[dn]}
                for dn in order]

    def _round_result(self, rec: Dict[str, Any]) -> Dict[str, Any]:
        """Shape a round record like the reference's logical_result row
        (run_task.py analyze_results:149-210)."""
        job = self.job
        return {
            "task_id": job.task_id,
            "logical_round": rec["round"] + 1,
            "logical_operator": self.last_operator,
            "logical_result": {"logical_result": self._result_entries(rec)},
            "round_failed": rec["round_failed"],
            "loss": rec["loss"],
        }
