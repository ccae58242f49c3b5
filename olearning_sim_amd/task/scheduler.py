"""Task scheduling: resource fit + priority scoring.

Parity with the reference's TaskScheduler (taskMgr/task_scheduler.py)
and DefaultStrategy (taskMgr/utils/scheduler_strategy.py):

- get_task_request_resource (scheduler_strategy.py:37-99): the
  logical-simulation demand is sum over resource_request tiers of
  (requested units x the tier's computation-unit num_cpus); mem uses
  the reference's per-unit default of 1.0; the device-simulation demand
  is the per-user phone counts.
- check_resource_availability (:101-148): logical cpu/mem fit AND every
  requested phone tier within the user's available quota.
- schedule_task (:150-161): score = queue-position term
  (len-i)/len + priority/10; highest score wins.
- TaskScheduler.freeze/release (task_scheduler.py:71-252) map onto the
  in-process ResourceManager.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from ..resource.manager import ResourceManager
from .queue import TaskQueue
from .schema import TaskConfig


@dataclass
class TaskSchedulerRes:
    task: TaskConfig
    task_request: Dict[str, Any]


class SchedulerStrategy:
    def schedule_next_task(self, task_queue: List[TaskConfig],
                           available: Dict[str, Any]) -> Optional[TaskSchedulerRes]:
        raise NotImplementedError


class DefaultStrategy(SchedulerStrategy):
    @staticmethod
    def get_task_request_resource(task: TaskConfig) -> Dict[str, Any]:
        unit_cpus = dict(zip(task.logical_simulation.computation_unit.devices,
                             [s.num_cpus for s in
                              task.logical_simulation.computation_unit.setting]))
        per_tier: Dict[str, int] = {}
        for rr in task.logical_simulation.resource_request:
            for tier, n in zip(rr.devices, rr.num_request):
                per_tier[tier] = per_tier.get(tier, 0) + n
        cpu = sum(unit_cpus.get(tier, 0) * n for tier, n in per_tier.items())
        mem = sum(1.0 * n for n in per_tier.values())  # reference default
        phones: Dict[str, int] = {}
        for rr in task.device_simulation.resource_request:
            for tier, n in zip(rr.devices, rr.num_request):
                phones[tier] = phones.get(tier, 0) + n
        return {"logical_simulation": {"cpu": float(cpu), "mem": float(mem)},
                "device_simulation": {task.user_id: phones} if phones else {}}

    @staticmethod
    def check_resource_availability(task_request: Dict[str, Any],
                                    available: Dict[str, Any]) -> bool:
        req = task_request.get("logical_simulation", {})
        avail = available.get("logical_simulation", {})
        if req.get("cpu", 0.0) > avail.get("cpu", 0.0):
            return False
        if req.get("mem", 0.0) > avail.get("mem", 0.0):
            return False
        for user, tiers in task_request.get("device_simulation", {}).items():
            user_avail = available.get("device_simulation", {}).get(user, {})
            for tier, n in tiers.items():
                if n > user_avail.get(tier, 0):
                    return False
        return True

    @staticmethod
    def schedule_task(waiting: List[Dict[str, Any]]) -> int:
        n = len(waiting)
        scores = [(n - i) / n + waiting[i]["task_priority"] / 10
                  for i in range(n)]
        return scores.index(max(scores))

    def schedule_next_task(self, task_queue: List[TaskConfig],
                           available: Dict[str, Any]) -> Optional[TaskSchedulerRes]:
        waiting = []
        for i, task in enumerate(task_queue):
            req = self.get_task_request_resource(task)
            if self.check_resource_availability(req, available):
                waiting.append({"index": i, "task": task, "request": req,
                                "task_priority": task.target.priority})
        if not waiting:
            return None
        pick = waiting[self.schedule_task(waiting)]
        return TaskSchedulerRes(task=pick["task"], task_request=pick["request"])


class StrategyFactory:
    _strategies = {"default": DefaultStrategy}

    @classmethod
    def create(cls, name: str = "default") -> SchedulerStrategy:
        return cls._strategies[name]()


class TaskScheduler:
    def __init__(self, resource_mgr: ResourceManager,
                 strategy: str = "default"):
        self.resources = resource_mgr
        self.strategy = StrategyFactory.create(strategy)

    def get_available_resources(self, user_id: str = "") -> Dict[str, Any]:
        return self.resources.get_resource(user_id)

    def run(self, queue: TaskQueue) -> Optional[TaskSchedulerRes]:
        tasks = queue.get_task_queue()
        if not tasks:
            return None
        # per-user phone availability folded per task at fit-check time
        available = self.resources.get_resource(
            tasks[0].user_id if tasks else "")
        for t in tasks[1:]:
            extra = self.resources.get_resource(t.user_id)
            available["device_simulation"].update(
                extra.get("device_simulation", {}))
        return self.strategy.schedule_next_task(tasks, available)

    def freeze(self, scheduled: TaskSchedulerRes) -> bool:
        req = scheduled.task_request
        phones = req.get("device_simulation", {}).get(
            scheduled.task.user_id, {})
        return self.resources.request_resource(
            task_id=scheduled.task.task_id,
            user_id=scheduled.task.user_id,
            cpu=req["logical_simulation"]["cpu"],
            mem=req["logical_simulation"]["mem"],
            phones=phones or None)

    def release(self, task_id: str) -> bool:
        return self.resources.release_resource(task_id)
