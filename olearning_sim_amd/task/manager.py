"""Task manager: lifecycle, scheduling loop, status fusion, watchdogs.

Parity with the reference's TaskManager
(ols_core/taskMgr/task_manager.py:51-1200):

- submitTask (:186-253): 3-stage validation -> duplicate/status checks
  against the table -> persist task_params + total_simulation ->
  status QUEUED -> enqueue.
- scheduler loop (`run`, :1053-1069): pick the next runnable task
  (TaskScheduler + DefaultStrategy), freeze resources, submit via
  TaskRunner, mark RUNNING + job_id + resource_occupied.
- releaseResource loop (:1071-1148): poll running tasks; when both
  sides finished, release quota, unregister deviceflow, stamp
  finish_task_time and the fused final status.
- interruptTask loop (:1150-1200): kill tasks queued longer than
  interrupt_queue_time or running longer than interrupt_running_time.
- status fusion combine_task_status / calculate_conditions
  (:610-889): the exact success/failure truth table over
  (logical_success, logical_round_failed, logical_job_status,
  device_success, device_round_failed, device_finished), with
  per-tier tolerance success >= nums - dynamic_nums and early round
  failure when failed > dynamic_nums.

Threads are opt-in (auto_start) — every loop body is also exposed as a
step_*() method so tests drive the state machine deterministically.
"""

from __future__ import annotations

import json
import threading
import time
from typing import Any, Dict, List, Optional, Tuple

from ..resource.manager import ResourceManager
from ..utils.logging import Logger
from .queue import TaskQueue
from .runner import TaskRunner
from .schema import TaskConfig, json2taskconfig
from .scheduler import TaskScheduler
from .status import TaskStatus, JobStatus
from .table import TaskTableRepo
from .validate import ValidateParameters

DEFAULT_TIMERS = {
    "scheduler_sleep_time": 5.0,
    "release_sleep_time": 10.0,
    "interrupt_sleep_time": 300.0,
    "interrupt_queue_time": 3600.0,
    "interrupt_running_time": 172800.0,
}


class TaskManager:
    def __init__(self, table: Optional[TaskTableRepo] = None,
                 resource_mgr: Optional[ResourceManager] = None,
                 runner: Optional[TaskRunner] = None,
                 timers: Optional[Dict[str, float]] = None,
                 deviceflow=None, auto_start: bool = False):
        self.table = table or TaskTableRepo(":memory:")
        self.resources = resource_mgr or ResourceManager(":memory:")
        self.runner = runner or TaskRunner(self.table)
        self.scheduler = TaskScheduler(self.resources)
        self.queue = TaskQueue()
        self.deviceflow = deviceflow
        self.timers = dict(DEFAULT_TIMERS, **(timers or {}))
        self.log = Logger.shared()
        self._running: List[str] = []   # task ids submitted & not released
        self._lock = threading.RLock()
        self._stopping = threading.Event()
        self._threads: List[threading.Thread] = []
        self._recover_from_repo()
        if auto_start:
            self.start()

    # -- crash recovery (reference get_taskqueue_from_repo, :89-155) ----
    def _recover_from_repo(self) -> None:
        for task_id in self.table.tasks_with_status(TaskStatus.QUEUED.value):
            params = self.table.get_item_value(task_id, "task_params")
            if params:
                try:
                    self.queue.add(json2taskconfig(params))
                except Exception:
                    self.table.set_item_value(task_id, "task_status",
                                              TaskStatus.FAILED.value)
        # tasks RUNNING when the previous process died are orphans: the
        # engine threads died with it and this runner knows nothing of
        # their job ids, so they would poll as RUNNING forever.  The
        # reference survives restarts because Ray tracks its jobs
        # (task_manager.py:521-524); here the job IS the process, so a
        # crashed run is a failed run.
        for task_id in self.table.tasks_with_status(TaskStatus.RUNNING.value):
            if task_id not in self.runner.task_jobs:
                self.table.set_items(task_id,
                                     task_status=TaskStatus.FAILED.value,
                                     resource_occupied=0,
                                     finish_task_time=time.time())
                self.resources.release_resource(task_id)
                self.log.error(task_id, "TaskMgr", "manager",
                               "orphaned RUNNING task failed on restart")
        # free resources frozen by a previous process with nothing running
        for task_id in self.resources.orphaned_tasks():
            occupied = self.table.get_item_value(task_id, "resource_occupied")
            status = self.table.get_item_value(task_id, "task_status")
            if status not in (TaskStatus.RUNNING.value,) or occupied != 1:
                self.resources.release_resource(task_id)

    # -- lifecycle -------------------------------------------------------
    def start(self) -> None:
        for target, period_key in ((self.step_schedule, "scheduler_sleep_time"),
                                   (self.step_release, "release_sleep_time"),
                                   (self.step_interrupt, "interrupt_sleep_time")):
            t = threading.Thread(target=self._loop,
                                 args=(target, self.timers[period_key]),
                                 daemon=True)
            t.start()
            self._threads.append(t)

    def shutdown(self) -> None:
        self._stopping.set()

    def _loop(self, fn, period: float) -> None:
        while not self._stopping.is_set():
            try:
                fn()
            except Exception as e:
                self.log.error("", "TaskMgr", "manager", f"loop error: {e}")
            self._stopping.wait(period)

    # -- submit ----------------------------------------------------------
    def submit_task(self, task_json: str) -> Tuple[bool, str]:
        try:
            raw = json.loads(task_json)
            cfg = json2taskconfig(task_json)
        except Exception as e:
            return False, f"invalid task JSON: {e}"
        v = ValidateParameters()
        if not v.validate_task_parameters(raw, cfg):
            return False, f"validation failed: {v.last_error}"
        task_id = cfg.task_id
        with self._lock:
            if self.queue.contains(task_id):
                return False, f"task {task_id} already queued"
            if self.table.has_task(task_id):
                status = self.table.get_item_value(task_id, "task_status")
                if status != TaskStatus.UNDONE.value and \
                        not TaskStatus(status).is_terminal():
                    return False, f"task {task_id} already active ({status})"
            else:
                self.table.add_task(task_id, user_id=cfg.user_id)
            self.table.set_items(
                task_id,
                task_status=TaskStatus.QUEUED.value,
                task_params=task_json,
                total_simulation=json.dumps(self._total_simulation(cfg)),
                in_queue_time=time.time(),
                resource_occupied=0, job_id=None,
                logical_target=None, device_target=None,
                logical_result=None, device_result=None,
                logical_round=None, logical_operator=None,
                device_round=None, device_operator=None,
                finish_task_time=None)
            self.queue.add(cfg)
        self.log.info(task_id, "TaskMgr", "manager", "task queued")
        return True, "queued"

    @staticmethod
    def _total_simulation(cfg: TaskConfig) -> Dict[str, Any]:
        """The `total_simulation` table item (task_manager.py:217-250)."""
        return {
            "max_round": cfg.operatorflow.flow_setting.round,
            "operator_name_list": [op.name for op in
                                   cfg.operatorflow.operators],
            "data_name_list": [d.name for d in cfg.target.data],
            "total_simulation": [
                {"name": d.name,
                 "simulation_target": {
                     "devices": list(d.total_simulation.devices),
                     "nums": list(d.total_simulation.nums),
                     "dynamic_nums": list(d.total_simulation.dynamic_nums)}}
                for d in cfg.target.data],
        }

    # -- scheduling step (reference run thread body) ---------------------
    def step_schedule(self) -> Optional[str]:
        with self._lock:
            scheduled = self.scheduler.run(self.queue)
            if scheduled is None:
                return None
            task = scheduled.task
            self.queue.remove(task.task_id)
            if not self.scheduler.freeze(scheduled):
                # put back (reference re-queues on freeze failure)
                self.queue.add(task)
                return None
            self.table.set_items(task.task_id, resource_occupied=1,
                                 freeze_time=time.time())
        job_id = None
        try:
            job_id = self.runner.submit(task)
        except Exception as e:
            self.log.error(task.task_id, "TaskMgr", "manager",
                           f"submit failed: {e}")
        with self._lock:
            if job_id is None and not self._has_device_side(task):
                self.scheduler.release(task.task_id)
                self.table.set_items(task.task_id,
                                     task_status=TaskStatus.FAILED.value,
                                     resource_occupied=0)
                return None
            self.table.set_items(task.task_id,
                                 task_status=TaskStatus.RUNNING.value,
                                 job_id=job_id,
                                 submit_task_time=time.time())
            self._running.append(task.task_id)
        return task.task_id

    @staticmethod
    def _has_device_side(task: TaskConfig) -> bool:
        return any(sum(rr.num_request) > 0
                   for rr in task.device_simulation.resource_request)

    # -- release step (reference releaseResource thread) ------------------
    def step_release(self) -> List[str]:
        released = []
        with self._lock:
            running = list(self._running)
        for task_id in running:
            status = self.get_task_status(task_id)
            if status.is_terminal():
                if self.deviceflow is not None:
                    try:
                        # reference releaseResource thread: resources are
                        # held until the gradient house has drained
                        # (utils.py check_deviceflow_dispatch_finished)
                        if not self.deviceflow.check_dispatch_finished(task_id):
                            continue
                        self.deviceflow.unregister_task(task_id)
                    except Exception:
                        pass
                self.scheduler.release(task_id)
                self.table.set_items(task_id,
                                     resource_occupied=0,
                                     finish_task_time=time.time(),
                                     release_time=time.time())
                with self._lock:
                    if task_id in self._running:
                        self._running.remove(task_id)
                released.append(task_id)
        return released

    # -- interrupt step (reference interruptTask thread) ------------------
    def step_interrupt(self) -> List[str]:
        interrupted = []
        now = time.time()
        for task_id in self.queue.get_task_ids():
            t0 = self.table.get_item_value(task_id, "in_queue_time") or now
            if now - t0 > self.timers["interrupt_queue_time"]:
                self.queue.remove(task_id)
                self.table.set_item_value(task_id, "task_status",
                                          TaskStatus.FAILED.value)
                interrupted.append(task_id)
        with self._lock:
            running = list(self._running)
        for task_id in running:
            t0 = self.table.get_item_value(task_id, "submit_task_time") or now
            if now - t0 > self.timers["interrupt_running_time"]:
                self.stop_task(task_id)
                interrupted.append(task_id)
        return interrupted

    # -- stop -------------------------------------------------------------
    def stop_task(self, task_id: str) -> Tuple[bool, str]:
        with self._lock:
            if self.queue.contains(task_id):
                self.queue.remove(task_id)
                self.table.set_item_value(task_id, "task_status",
                                          TaskStatus.STOPPED.value)
                return True, "removed from queue"
        if self.table.get_item_value(task_id, "resource_occupied") == 1:
            self.runner.stop_task(task_id)
            return True, "stop requested"
        return False, "task not queued or running"

    # -- status fusion -----------------------------------------------------
    def change_scheduler(self, name: str) -> bool:
        """Swap the scheduling strategy by name (reference changeScheduler
        RPC, taskService.proto; StrategyFactory holds the registry)."""
        from .scheduler import StrategyFactory
        try:
            self.scheduler.strategy = StrategyFactory.create(name or "default")
            return True
        except KeyError:
            return False

    def get_task_queue(self) -> List[str]:
        return self.queue.get_task_ids()

    def get_task_status(self, task_id: str) -> TaskStatus:
        if self.queue.contains(task_id):
            return TaskStatus.QUEUED
        row = self.table.get_row(task_id)
        if row is None:
            return TaskStatus.MISSING
        if row.get("resource_occupied") == 1:
            job_id = row.get("job_id")
            logical_status = None
            if job_id:
                js = self.runner.get_job_status(job_id)
                logical_status = TaskStatus(js.value) if js else None
            device_result = self._device_task_result(task_id, row)
            fused = self.combine_task_status(task_id, logical_status,
                                             device_result)
            if fused.is_terminal():
                self.table.set_item_value(task_id, "task_status", fused.value)
            return fused
        status = row.get("task_status")
        return TaskStatus(status) if status else TaskStatus.MISSING

    def _device_task_result(self, task_id: str, row: Dict[str, Any]
                            ) -> Dict[str, Any]:
        """Build the reference's DeviceTaskResult dict from the simulated
        phone farm (task_manager.py:537-576)."""
        result = {"is_finished": True, "device_result": []}
        if row.get("device_target"):
            dr = row.get("device_result")
            result["device_result"] = (json.loads(dr)["device_result"]
                                       if dr else [])
            finished = False
            # only THIS task's simulated phone job (runner.task_jobs)
            for jid in self.runner.task_jobs.get(task_id, []):
                h = self.runner.jobs.get(jid)
                if h is not None and h.kind == "device":
                    finished = h.status in (JobStatus.SUCCEEDED,
                                            JobStatus.FAILED,
                                            JobStatus.STOPPED)
            result["is_finished"] = finished
        return result

    # reference combine_task_status (task_manager.py:610-697)
    def combine_task_status(self, task_id: str,
                            logical_task_status: Optional[TaskStatus],
                            device_task_result: Dict[str, Any]) -> TaskStatus:
        device_finished = device_task_result.get("is_finished", True)
        ts_string = self.table.get_item_value(task_id, "total_simulation")
        if ts_string is None:
            return TaskStatus.FAILED
        task_params = json.loads(ts_string)

        (logical_success, logical_round_failed,
         device_success, device_round_failed) = self.calculate_conditions(
            task_id, task_params, device_task_result)

        if logical_success and logical_round_failed:
            return TaskStatus.FAILED
        if device_success and device_round_failed:
            return TaskStatus.FAILED
        if logical_success and device_success:
            return TaskStatus.SUCCEEDED
        if (not logical_success and not logical_round_failed
                and logical_task_status == TaskStatus.STOPPED
                and not device_round_failed and device_finished):
            return TaskStatus.STOPPED
        if not logical_success and logical_task_status in (
                TaskStatus.SUCCEEDED, TaskStatus.FAILED, TaskStatus.STOPPED):
            return TaskStatus.FAILED
        if not logical_success and logical_round_failed:
            return TaskStatus.FAILED
        if not device_success and device_finished:
            return TaskStatus.FAILED
        if not device_success and device_round_failed:
            return TaskStatus.FAILED
        return TaskStatus.RUNNING

    # reference calculate_conditions (task_manager.py:699-889)
    def calculate_conditions(self, task_id: str, task_params: Dict[str, Any],
                             device_task_result: Dict[str, Any]):
        max_round = task_params.get("max_round", 0)
        operator_names = task_params.get("operator_name_list", [])
        data_names = task_params.get("data_name_list", [])
        total_simulation = task_params.get("total_simulation", [])
        last_operator = operator_names[-1] if operator_names else ""

        def success_check(target, result, result_names, current_round,
                          operator_name):
            """last round + last operator + per-tier success tolerance"""
            oks = []
            if current_round is not None and current_round >= max_round \
                    and operator_name == last_operator:
                for idx, data_total in enumerate(total_simulation):
                    name = data_names[idx]
                    tgt_nums = target[idx].get("simulation_target", {}) \
                        .get("nums", []) if idx < len(target) else []
                    dyn = data_total.get("simulation_target", {}) \
                        .get("dynamic_nums", [])
                    if name in result_names:
                        ridx = result_names.index(name)
                        succ = result[ridx].get("simulation_target", {}) \
                            .get("success_num", [])
                        # every targeted tier must be reported AND meet
                        # its tolerance — a shorter vector is a miss
                        oks.append(len(succ) >= len(tgt_nums)
                                   and all(x >= a - z for x, a, z in
                                           zip(succ, tgt_nums, dyn)))
                    elif tgt_nums:
                        # a targeted data with no result entry at all
                        # cannot be a success
                        oks.append(False)
            return bool(oks) and all(oks)

        # logical side
        lt_string = self.table.get_item_value(task_id, "logical_target")
        logical_result, logical_names = [], []
        logical_round = logical_operator = None
        if lt_string is not None:
            logical_success = logical_round_failed = False
            lr_string = self.table.get_item_value(task_id, "logical_result")
            if lr_string is not None:
                logical_target = json.loads(lt_string).get("logical_target", [])
                logical_result = json.loads(lr_string).get("logical_result", [])
                logical_round = self.table.get_item_value(task_id,
                                                          "logical_round")
                logical_operator = self.table.get_item_value(
                    task_id, "logical_operator")
                logical_names = [r.get("name", "") for r in logical_result]
                if success_check(logical_target, logical_result,
                                 logical_names, logical_round,
                                 logical_operator):
                    logical_success, logical_round_failed = True, False
        else:
            logical_success, logical_round_failed = True, False

        # device side
        dt_string = self.table.get_item_value(task_id, "device_target")
        device_result, device_names = [], []
        device_round = device_operator = None
        if dt_string is not None:
            device_success = device_round_failed = False
            if device_task_result.get("device_result", []):
                device_target = json.loads(dt_string).get("device_target", [])
                device_result = device_task_result.get("device_result", [])
                device_round = self.table.get_item_value(task_id,
                                                         "device_round")
                device_operator = self.table.get_item_value(task_id,
                                                            "device_operator")
                device_names = [r.get("name", "") for r in device_result]
                if success_check(device_target, device_result, device_names,
                                 device_round, device_operator):
                    device_success, device_round_failed = True, False
        else:
            device_success, device_round_failed = True, False

        # combined per-data round-failure / combined-success checks
        combine_status: List[bool] = []
        for idx, data_total in enumerate(total_simulation):
            name = data_names[idx]
            nums = data_total.get("simulation_target", {}).get("nums", [])
            dyn = data_total.get("simulation_target", {}).get("dynamic_nums", [])
            if not dyn:
                dyn = [0] * len(nums)
            def _pad(v, n):
                # a result vector shorter than the target counts its
                # missing tiers as 0 successes (never zip-truncates)
                return list(v) + [0] * max(0, n - len(v))

            if name in logical_names:
                li = logical_names.index(name)
                lf = _pad(logical_result[li].get("simulation_target", {})
                          .get("failed_num", []), len(dyn))
                ls = _pad(logical_result[li].get("simulation_target", {})
                          .get("success_num", []), len(nums))
            else:
                lf, ls = [0] * len(dyn), [0] * len(nums)
            if name in device_names:
                di = device_names.index(name)
                df = _pad(device_result[di].get("simulation_target", {})
                          .get("failed_num", []), len(dyn))
                ds = _pad(device_result[di].get("simulation_target", {})
                          .get("success_num", []), len(nums))
            else:
                df, ds = [0] * len(dyn), [0] * len(nums)

            failed_cmp: List[bool] = []
            if logical_result == [] or device_result == []:
                failed_cmp = [z < x + y for z, x, y in zip(dyn, lf, df)]
            if (logical_round is not None and device_round is not None
                    and logical_round == device_round
                    and logical_operator == device_operator):
                failed_cmp = [z < x + y for z, x, y in zip(dyn, lf, df)]
            if failed_cmp and any(failed_cmp):
                if logical_result == [] and device_result != []:
                    logical_round_failed, device_round_failed = False, True
                elif logical_result != [] and device_result == []:
                    logical_round_failed, device_round_failed = True, False
                else:
                    logical_round_failed = device_round_failed = True
                break

            success_cmp: List[bool] = []
            if logical_result == [] or device_result == []:
                success_cmp = [x + y >= a - z for x, y, a, z in
                               zip(ls, ds, nums, dyn)]
            if (logical_round is not None and device_round is not None
                    and logical_round == device_round):
                success_cmp = [x + y >= a - z for x, y, a, z in
                               zip(ls, ds, nums, dyn)]
            if success_cmp:
                combine_status.append(all(success_cmp))

        if logical_result:
            if logical_round is not None and logical_round >= max_round \
                    and logical_operator == last_operator:
                if combine_status and all(combine_status):
                    logical_success = True
        if device_result:
            if device_round is not None and device_round >= max_round \
                    and device_operator == last_operator:
                if combine_status and all(combine_status):
                    device_success = True

        return (logical_success, logical_round_failed,
                device_success, device_round_failed)
