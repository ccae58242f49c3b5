"""Task runner: turn a scheduled TaskConfig into running simulations.

Parity with the reference's TaskRunner (taskMgr/task_runner.py:30-256):
`submit` computes the hybrid allocation, assembles the per-side job
descriptions, registers with deviceflow when the gradient house is in
use, writes logical_target/device_target rows, and launches

- the logical simulation: an in-process LogicalEngine thread (the
  reference submits run_task.py as a Ray job, task_runner.py:41-87;
  here the engine IS the execution plane), and
- the device simulation: a PhoneFarmSimulator thread that models the
  real-phone farm's completion using the reference's published cost
  model (LAMBDA + BETA * nums / phones) — the proprietary PhoneMgr farm
  is not reachable, so its *timing and result semantics* are simulated.

Operator knobs come from the train operator's `operator_params` JSON
(the operator-facing API of taskMgr/base/base_operator.py:12-53):
model, lr, local_steps, batch_size, prox_mu, cohort_size, num_classes,
dirichlet_alpha, dtype, chunk_clients, vocab_size, seq_len.
"""

from __future__ import annotations

import os
import json
import threading
import time
import uuid
from typing import Any, Dict, List, Optional

from ..engine.job import EngineJob
from ..utils.logging import Logger
from .allocation import HybridOptimizer, DataAllocation, BETA, LAMBDA
from .schema import TaskConfig
from .submitter import JobSubmitter, fix_device_task_json, json2deviceconfig
from .status import JobStatus
from .table import TaskTableRepo


class JobHandle:
    """Tracks one side's execution (reference: Ray job id / phone task)."""

    def __init__(self, job_id: str, kind: str):
        self.job_id = job_id
        self.kind = kind
        self.status = JobStatus.PENDING
        self.error: Optional[str] = None
        self._stop = threading.Event()
        self.thread: Optional[threading.Thread] = None

    def request_stop(self) -> None:
        self._stop.set()

    @property
    def stop_requested(self) -> bool:
        return self._stop.is_set()


def engine_job_from_task(task: TaskConfig, allocations: List[DataAllocation],
                         device: str = "cpu",
                         checkpoint_dir: str = "") -> EngineJob:
    """Build the logical-simulation EngineJob from the task config."""
    train_op = None
    for op in task.operatorflow.operators:
        if op.logical_simulation.operator_code_path or \
                op.logical_simulation.operator_entry_file:
            train_op = op
            break
    params: Dict[str, Any] = {}
    if train_op is not None and train_op.logical_simulation.operator_params:
        try:
            params = json.loads(train_op.logical_simulation.operator_params)
        except Exception:
            params = {}

    clients = sum(a.logical_total() for a in allocations)
    dynamic = sum(sum(d.total_simulation.dynamic_nums)
                  for d in task.target.data)
    behavior = ""
    if train_op is not None and \
            train_op.operation_behavior_controller.use_gradient_house:
        behavior = train_op.operation_behavior_controller.strategy_gradient_house

    # ordered operator list for the round loop: kind from each
    # operator's params JSON ("kind") or inferred from its name
    op_entries = []
    for op in task.operatorflow.operators:
        if not (op.logical_simulation.operator_code_path
                or op.logical_simulation.operator_entry_file):
            continue
        try:
            op_params = json.loads(op.logical_simulation.operator_params or "{}")
        except Exception:
            op_params = {}
        kind = op_params.get("kind")
        if kind is None:
            code = op.logical_simulation.operator_code_path
            nm = op.name.lower()
            if code and not code.startswith("builtin:"):
                kind = "script"     # user script-file operator
            elif "eval" in nm:
                kind = "evaluate"
            elif "checkpoint" in nm or "save" in nm:
                kind = "checkpoint"
            else:
                kind = "train"
        op_entries.append((op.name, kind))

    fs = task.operatorflow.flow_setting
    first_data = task.target.data[0] if task.target.data else None
    job = EngineJob(
        operators=op_entries or [("train", "train")],
        flow_start_strategy=fs.start.logical_simulation.strategy,
        flow_stop_strategy=fs.stop.logical_simulation.strategy,
        flow_wait_interval=max(1, fs.stop.logical_simulation.wait_interval
                               or fs.start.logical_simulation.wait_interval or 1),
        flow_total_timeout=max(fs.start.logical_simulation.total_timeout,
                               fs.stop.logical_simulation.total_timeout),
        task_id=task.task_id,
        model_name=params.get("model", "mlp"),
        model_kwargs=params.get("model_kwargs", {}),
        clients=max(1, clients),
        cohort_size=params.get("cohort_size", 0),
        rounds=max(1, task.operatorflow.flow_setting.round),
        local_steps=params.get("local_steps", 2),
        batch_size=params.get("batch_size", 8),
        lr=params.get("lr", 0.05),
        prox_mu=params.get("prox_mu", 0.0),
        dtype=params.get("dtype", "float32" if device == "cpu" else "bfloat16"),
        device=device,
        chunk_clients=params.get("chunk_clients", 0),
        seed=params.get("seed", 1234),
        num_classes=params.get("num_classes", 10),
        dirichlet_alpha=params.get("dirichlet_alpha", 0.1),
        shard_size=params.get("shard_size", 64),
        vocab_size=params.get("vocab_size", 0),
        seq_len=params.get("seq_len", 0),
        behavior_strategy=behavior,
        # crash recovery of an interrupted task is opt-in via the
        # operator_params JSON (resume: true) — a fresh submission with
        # the same task_id must NOT silently load stale artifacts
        resume=bool(params.get("resume", False)),
        checkpoint_dir=checkpoint_dir,
        save_every_round=bool(train_op and train_op.model.use_model
                              and checkpoint_dir),
        model_update_style=(train_op.model.model_update_style
                            if train_op else ""),
        data_name=first_data.name if first_data else "data_0",
        device_tier=(first_data.total_simulation.devices[0]
                     if first_data and first_data.total_simulation.devices
                     else "high"),
        dynamic_num=dynamic,
        tier_counts=[(t.tier, t.logical) for a in allocations[:1]
                     for t in a.tiers],
        data_segments=[(a.data_name, t.tier, t.logical)
                       for a in allocations for t in a.tiers
                       if t.logical > 0],
        dynamic_nums=_segment_dynamic_nums(task, allocations),
    )
    return job


def _segment_dynamic_nums(task: TaskConfig,
                          allocations: List[DataAllocation]) -> List[int]:
    """Per-(data x tier) failure tolerance aligned with data_segments."""
    by_data = {d.name: d for d in task.target.data}
    out: List[int] = []
    for a in allocations:
        d = by_data.get(a.data_name)
        dyn = list(d.total_simulation.dynamic_nums) if d else []
        devs = list(d.total_simulation.devices) if d else []
        for t in a.tiers:
            if t.logical <= 0:
                continue
            try:
                out.append(dyn[devs.index(t.tier)])
            except (ValueError, IndexError):
                out.append(0)
    return out


class TaskRunner:
    def __init__(self, table: TaskTableRepo, device: str = "cpu",
                 checkpoint_dir: str = "", deviceflow=None, perf=None,
                 cluster=None):
        self.table = table
        self.device = device
        self.checkpoint_dir = checkpoint_dir
        self.deviceflow = deviceflow   # deviceflow service facade (optional)
        self.perf = perf               # PerformanceManager (optional)
        # cluster: NodeClusterManager — tasks whose train operator asks
        # for num_gpus > 1 run as a one-process-per-GPU worker group
        # (reference TaskRunner submits to the Ray cluster fabric,
        # task_runner.py:41-87)
        self.cluster = cluster
        self.jobs: Dict[str, JobHandle] = {}
        self.task_jobs: Dict[str, List[str]] = {}   # task_id -> job ids
        self.log = Logger.shared()
        self._lock = threading.Lock()

    # ------------------------------------------------------------------
    def submit(self, task: TaskConfig) -> Optional[str]:
        """Allocate, register, launch.  Returns the logical job id."""
        allocations = HybridOptimizer(task).allocate()

        logical_target = {"logical_target": [
            {"name": a.data_name,
             "simulation_target": {
                 "devices": [t.tier for t in a.tiers],
                 "nums": [t.logical for t in a.tiers],
                 "dynamic_nums": []}}
            for a in allocations if a.logical_total() > 0]}
        device_target = {"device_target": [
            {"name": a.data_name,
             "simulation_target": {
                 "devices": [t.tier for t in a.tiers],
                 "nums": [t.device for t in a.tiers],
                 "dynamic_nums": []}}
            for a in allocations if a.device_total() > 0]}

        if logical_target["logical_target"]:
            self.table.set_item_value(task.task_id, "logical_target",
                                      json.dumps(logical_target))
        if device_target["device_target"]:
            self.table.set_item_value(task.task_id, "device_target",
                                      json.dumps(device_target))

        # per-side assembled task JSONs (reference JobSubmitter,
        # utils_runner.py:478-628) — persisted so the exact config each
        # side ran under is auditable from the task table
        sub = JobSubmitter(task, allocations)
        side_logical = sub.assemble_info_logical_simulation()
        if side_logical is not None:
            self.table.set_item_value(task.task_id, "logical_task_params",
                                      json.dumps(side_logical))
        side_device = sub.assemble_info_device_simulation()
        if side_device is not None:
            device_cfg = fix_device_task_json(
                json2deviceconfig(side_device), task.task_id)
            self.table.set_item_value(task.task_id, "device_task_params",
                                      json.dumps(device_cfg))

        if self.deviceflow is not None:
            for op in task.operatorflow.operators:
                if op.operation_behavior_controller.use_gradient_house:
                    self.deviceflow.register_task(
                        task.task_id, total_compute_resources=self._sides(
                            allocations))
                    break

        job_id = None
        if logical_target["logical_target"]:
            job_id = self._submit_logical(task, allocations)
        if device_target["device_target"]:
            self._submit_phone(task, allocations)
        return job_id

    def _stage_script_operators(self, task: TaskConfig, job: EngineJob):
        """Stage user script-file operator code and build executors
        (reference get_operator_code + Actor, utils_runner.py:684-782,
        utils_run_task.py:146-577)."""
        script_names = {name for name, kind in job.operators
                        if kind == "script"}
        if not script_names:
            return {}
        import tempfile
        from ..engine.script_op import ScriptOperator
        from .staging import stage_operator_code
        work_root = os.path.join(self.checkpoint_dir or tempfile.gettempdir(),
                                 f"opwork_{task.task_id}")
        first_data = task.target.data[0] if task.target.data else None
        data_info = ({"name": first_data.name,
                      "data_path": first_data.data_path,
                      "data_split_type": first_data.data_split_type,
                      "task_type": first_data.task_type}
                     if first_data else {})
        ops = {}
        for op in task.operatorflow.operators:
            if op.name not in script_names:
                continue
            sim = op.logical_simulation
            staged = stage_operator_code(
                sim.operator_code_path, sim.operator_entry_file, op.name,
                work_root)
            model_info = {
                "use_model": op.model.use_model,
                "model_path": op.model.model_path,
                "model_update_style": op.model.model_update_style,
            }
            ops[op.name] = ScriptOperator(
                name=op.name, staged_dir=staged,
                entry_file=sim.operator_entry_file,
                operator_params=sim.operator_params,
                task_id=task.task_id, work_dir=work_root,
                clients=job.clients,
                shards=min(8, max(1, job.clients // 4)),
                data_info=data_info, model_info=model_info)
        return ops

    @staticmethod
    def _sides(allocations: List[DataAllocation]) -> List[str]:
        sides = []
        if any(a.logical_total() > 0 for a in allocations):
            sides.append("logical_simulation")
        if any(a.device_total() > 0 for a in allocations):
            sides.append("device_simulation")
        return sides

    @staticmethod
    def _group_world_size(task: TaskConfig) -> int:
        """Worker-group size from the train operator's params JSON
        (operator-facing knob ``num_gpus``; 0/1 = in-process)."""
        for op in task.operatorflow.operators:
            try:
                params = json.loads(op.logical_simulation.operator_params
                                    or "{}")
            except Exception:
                continue
            n = params.get("num_gpus", 0)
            if isinstance(n, int) and n > 1:
                return n
        return 1

    # -- logical side ----------------------------------------------------
    def _submit_logical(self, task: TaskConfig,
                        allocations: List[DataAllocation]) -> str:
        job = engine_job_from_task(task, allocations, self.device,
                                   self.checkpoint_dir)
        world = self._group_world_size(task)
        has_script = any(kind == "script" for _, kind in job.operators)
        if world > 1 and self.cluster is not None and not has_script:
            return self._submit_logical_group(task, job, world)
        script_ops = self._stage_script_operators(task, job)
        job_id = f"olsjob_{uuid.uuid4().hex[:12]}"
        handle = JobHandle(job_id, "logical")

        def run():
            handle.status = JobStatus.RUNNING
            try:
                from ..engine.round_loop import LogicalEngine

                def sink(row: Dict[str, Any]) -> None:
                    self.table.set_items(
                        task.task_id,
                        logical_round=row["logical_round"],
                        logical_operator=row["logical_operator"],
                        logical_result=json.dumps(row["logical_result"]))

                eng = LogicalEngine(job, result_sink=sink,
                                    deviceflow=self.deviceflow,
                                    perf=self.perf, script_ops=script_ops)
                # cooperative stop (reference: JobSubmissionClient.stop_job)
                orig_run_round = eng.run_round

                def run_round(r):
                    if handle.stop_requested:
                        eng.stop_requested = True
                    return orig_run_round(r)

                eng.run_round = run_round
                out = eng.run()
                if handle.stop_requested:
                    handle.status = JobStatus.STOPPED
                elif any(rec.get("round_failed") for rec in out["records"]):
                    handle.status = JobStatus.FAILED
                else:
                    handle.status = JobStatus.SUCCEEDED
            except Exception as e:  # engine crash -> FAILED, like a Ray job
                handle.error = str(e)
                handle.status = JobStatus.FAILED
                self.log.error(task.task_id, "TaskMgr", "runner",
                               f"logical job failed: {e}")

        handle.thread = threading.Thread(target=run, daemon=True)
        with self._lock:
            self.jobs[job_id] = handle
            self.task_jobs.setdefault(task.task_id, []).append(job_id)
        handle.thread.start()
        return job_id

    def _submit_logical_group(self, task: TaskConfig, job: EngineJob,
                              world: int) -> str:
        """Launch the logical simulation as a one-process-per-GPU worker
        group through the NodeClusterManager (the execution fabric; one
        rank per GPU over RCCL, gloo on CPU).  A monitor thread polls
        the group and replays rank 0's per-round result rows into the
        task table on completion, so status fusion sees the same
        logical_round/logical_result rows the in-process path writes."""
        import dataclasses
        import tempfile

        if self.device.startswith("cuda"):
            import torch
            avail = torch.cuda.device_count()
            if avail > 0:
                world = min(world, avail)

        job_id = f"olsgrp_{uuid.uuid4().hex[:12]}"
        handle = JobHandle(job_id, "logical")
        work_dir = os.path.join(self.checkpoint_dir or tempfile.gettempdir(),
                                f"group_{task.task_id}_{job_id}")
        os.makedirs(work_dir, exist_ok=True)
        spec_path = os.path.join(work_dir, "job.json")
        result_path = os.path.join(work_dir, "result.json")
        spec = dataclasses.asdict(job)
        spec["device"] = ""           # each rank picks cuda:LOCAL_RANK / cpu
        with open(spec_path, "w") as f:
            json.dump(spec, f)

        from ..cluster.node_manager import WorkerGroupSpec
        cname = f"grp_{job_id}"
        created = self.cluster.create_cluster(WorkerGroupSpec(
            name=cname, replicas=world,
            entry_module="olearning_sim_amd.engine.worker",
            args=["--job-json", spec_path, "--result-json", result_path]))
        if not created:
            handle.status = JobStatus.FAILED
            handle.error = f"worker group {cname} already exists"
            with self._lock:
                self.jobs[job_id] = handle
                self.task_jobs.setdefault(task.task_id, []).append(job_id)
            return job_id

        def monitor():
            handle.status = JobStatus.RUNNING
            try:
                while True:
                    info = self.cluster.get_cluster(cname)
                    state = info["status"] if info else "failed"
                    if handle.stop_requested:
                        self.cluster.delete_cluster(cname)
                        handle.status = JobStatus.STOPPED
                        return
                    if state in ("succeeded", "failed"):
                        break
                    time.sleep(0.3)
                out = None
                if os.path.exists(result_path):
                    try:
                        with open(result_path) as f:
                            out = json.load(f)
                    except Exception:
                        out = None
                if out is not None:
                    for row in out.get("round_rows", []):
                        self.table.set_items(
                            task.task_id,
                            logical_round=row["logical_round"],
                            logical_operator=row["logical_operator"],
                            logical_result=json.dumps(row["logical_result"]))
                    failed = any(r.get("round_failed")
                                 for r in out.get("round_rows", []))
                else:
                    failed = True
                if state == "succeeded" and not failed:
                    handle.status = JobStatus.SUCCEEDED
                else:
                    handle.status = JobStatus.FAILED
                    if out is None:
                        handle.error = "worker group produced no result"
                self.cluster.delete_cluster(cname)
            except Exception as e:
                handle.error = str(e)
                handle.status = JobStatus.FAILED
                self.log.error(task.task_id, "TaskMgr", "runner",
                               f"worker group failed: {e}")

        handle.thread = threading.Thread(target=monitor, daemon=True)
        with self._lock:
            self.jobs[job_id] = handle
            self.task_jobs.setdefault(task.task_id, []).append(job_id)
        handle.thread.start()
        self.log.info(task.task_id, "TaskMgr", "runner",
                      f"logical simulation on worker group {cname} "
                      f"(world={world})")
        return job_id

    # -- device side (simulated phone farm) -----------------------------
    def _submit_phone(self, task: TaskConfig,
                      allocations: List[DataAllocation]) -> str:
        job_id = f"phone_{uuid.uuid4().hex[:12]}"
        handle = JobHandle(job_id, "device")
        rounds = max(1, task.operatorflow.flow_setting.round)
        phones_map: Dict[str, int] = {}
        for rr in task.device_simulation.resource_request:
            for tier, n in zip(rr.devices, rr.num_request):
                phones_map[tier] = phones_map.get(tier, 0) + n

        def run():
            handle.status = JobStatus.RUNNING
            try:
                for r in range(rounds):
                    per_round = 0.0
                    for a in allocations:
                        for t in a.tiers:
                            if t.device > 0:
                                per_round = max(
                                    per_round,
                                    LAMBDA + BETA * t.device
                                    / max(1, phones_map.get(t.tier, 1)))
                    # scaled-down wait: the farm timing model, compressed
                    # so simulated phone rounds do not dominate tests
                    waited = 0.0
                    while waited < min(per_round * 0.01, 2.0):
                        if handle.stop_requested:
                            handle.status = JobStatus.STOPPED
                            return
                        time.sleep(0.01)
                        waited += 0.01
                    result = {"device_result": [
                        {"name": a.data_name,
                         "simulation_target": {
                             "devices": [t.tier for t in a.tiers],
                             "success_num": [t.device for t in a.tiers],
                             "failed_num": [0 for _ in a.tiers]}}
                        for a in allocations if a.device_total() > 0]}
                    self.table.set_items(
                        task.task_id,
                        device_round=r + 1,
                        device_operator="train",
                        device_result=json.dumps(result))
                handle.status = JobStatus.SUCCEEDED
            except Exception as e:
                handle.error = str(e)
                handle.status = JobStatus.FAILED

        handle.thread = threading.Thread(target=run, daemon=True)
        with self._lock:
            self.jobs[job_id] = handle
            self.task_jobs.setdefault(task.task_id, []).append(job_id)
        handle.thread.start()
        return job_id

    # ------------------------------------------------------------------
    def get_job_status(self, job_id: str) -> Optional[JobStatus]:
        h = self.jobs.get(job_id)
        return h.status if h else None

    def stop_job(self, job_id: str) -> bool:
        h = self.jobs.get(job_id)
        if h is None:
            return False
        h.request_stop()
        return True

    def stop_task(self, task_id: str) -> None:
        """Stop every job (logical + simulated device side) this task
        launched — and ONLY this task's (reference stopTask stops the
        Ray job by id and the phone task by task_id)."""
        with self._lock:
            ids = list(self.task_jobs.get(task_id, []))
        job_id = self.table.get_item_value(task_id, "job_id")
        if job_id and job_id not in ids:
            ids.append(job_id)
        for jid in ids:
            self.stop_job(jid)

    def wait(self, job_id: str, timeout: float = 60.0) -> Optional[JobStatus]:
        h = self.jobs.get(job_id)
        if h is None:
            return None
        if h.thread is not None:
            h.thread.join(timeout)
        return h.status
