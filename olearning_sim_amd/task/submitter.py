"""Per-side task assembly and device-task conversion.

Parity with the reference's JobSubmitter
(ols_core/taskMgr/utils/utils_runner.py:478-628) and the device-task
converters (`json2deviceconfig` :784-902, `fix_device_task_json`
:905-932): once the hybrid allocation has split each data's
machine-times between the logical-simulation and device-simulation
sides, each side receives its OWN task JSON — data paths swapped for
the side's split, `total_simulation` rewritten to the side's numbers,
the other side's per-operator code/params stripped — and the device
side is further lowered to the flat config a phone farm consumes.
"""

from __future__ import annotations

import copy
import json
from typing import Any, Dict, List, Optional

from .allocation import DataAllocation
from .schema import TaskConfig, taskconfig2json


def _side_path(path: str, side: str) -> str:
    """Split-data path naming: `<stem>_<side><ext>` — the analogue of
    the reference splitter's re-uploaded `logical_/device_` zips
    (utils_runner.py:195-327)."""
    if not path:
        return path
    if path.endswith(".zip"):
        return path[:-4] + f"_{side}.zip"
    return path.rstrip("/") + f"_{side}"


def _scale_dynamic(dyn: int, part: int, total: int) -> int:
    """Failure tolerance follows the population proportionally
    (floor), so the two sides together never tolerate more than the
    original budget."""
    if total <= 0:
        return 0
    return (dyn * part) // total


class JobSubmitter:
    """Assemble the side-specific task configs from the allocation."""

    def __init__(self, task: TaskConfig, allocations: List[DataAllocation]):
        self.task = task
        self.alloc = {a.data_name: a for a in allocations}

    # -- shared -----------------------------------------------------------
    def _assemble(self, side: str) -> Optional[Dict[str, Any]]:
        """side is 'logical' or 'device'."""
        other = "device" if side == "logical" else "logical"
        cfg = copy.deepcopy(self.task)
        any_work = False
        kept_data = []
        for d in cfg.target.data:
            a = self.alloc.get(d.name)
            if a is None:
                continue
            nums = [t.logical if side == "logical" else t.device
                    for t in a.tiers]
            if sum(nums) == 0:
                continue    # this data has no work on this side
            any_work = True
            totals = [t.total for t in a.tiers]
            d.data_path = _side_path(d.data_path, side)
            dyn = list(d.total_simulation.dynamic_nums)
            d.total_simulation.devices = [t.tier for t in a.tiers]
            d.total_simulation.nums = nums
            d.total_simulation.dynamic_nums = [
                _scale_dynamic(dyn[i] if i < len(dyn) else 0, nums[i],
                               totals[i]) for i in range(len(nums))]
            # the split is already decided: the side JSON carries no
            # further allocation freedom
            d.allocation.optimization = False
            d.allocation.logical_simulation = (
                nums if side == "logical" else [0] * len(nums))
            d.allocation.device_simulation = (
                nums if side == "device" else [0] * len(nums))
            kept_data.append(d)
        if not any_work:
            return None
        cfg.target.data = kept_data
        for op in cfg.operatorflow.operators:
            # strip the other side's operator info
            # (assemble_info_*_simulation, utils_runner.py:498-628)
            from .schema import OperatorSimulationInfo
            setattr(op, f"{other}_simulation", OperatorSimulationInfo())
        return json.loads(taskconfig2json(cfg))

    def assemble_info_logical_simulation(self) -> Optional[Dict[str, Any]]:
        return self._assemble("logical")

    def assemble_info_device_simulation(self) -> Optional[Dict[str, Any]]:
        return self._assemble("device")


# -- device-task conversion (phone-farm side) -----------------------------

def json2deviceconfig(task_json: Dict[str, Any]) -> Dict[str, Any]:
    """Lower the device-side task JSON to the flat config a phone farm
    consumes (reference json2deviceconfig, utils_runner.py:784-902):
    per-data populations, per-operator entry points, model info, and
    the round count — no allocation or logical-side structure."""
    flow = task_json.get("operatorflow", {})
    data = []
    for d in task_json.get("target", {}).get("data", []):
        ts = d.get("total_simulation", {})
        data.append({
            "name": d.get("name", ""),
            "data_path": d.get("data_path", ""),
            "data_transfer_type": d.get("data_transfer_type", ""),
            "task_type": d.get("task_type", ""),
            "devices": list(ts.get("devices", [])),
            "nums": list(ts.get("nums", [])),
        })
    operators = []
    for op in flow.get("operators", []):
        dev = op.get("device_simulation", {})
        operators.append({
            "name": op.get("name", ""),
            "use_data": op.get("use_data", False),
            "entry_file": dev.get("operator_entry_file", ""),
            "code_path": dev.get("operator_code_path", ""),
            "transfer_type": dev.get("operator_transfer_type", ""),
            "params": dev.get("operator_params", ""),
            "use_gradient_house": op.get(
                "operation_behavior_controller", {}).get(
                "use_gradient_house", False),
            "outbound_service": op.get(
                "operation_behavior_controller", {}).get(
                "outbound_service", ""),
            "model": dict(op.get("model", {})),
        })
    return {
        "task_id": task_json.get("task_id", ""),
        "user_id": task_json.get("user_id", ""),
        "rounds": flow.get("flow_setting", {}).get("round", 0),
        "data": data,
        "operators": operators,
    }


def fix_device_task_json(device_cfg: Dict[str, Any], task_id: str,
                         outbound_url: str = "") -> Dict[str, Any]:
    """Inject the task id into every operator's params and rewrite the
    outbound service to the concrete endpoint
    (fix_device_task_json, utils_runner.py:905-932)."""
    cfg = copy.deepcopy(device_cfg)
    cfg["task_id"] = task_id
    for op in cfg.get("operators", []):
        try:
            params = json.loads(op.get("params") or "{}")
        except (ValueError, TypeError):
            params = {}
        params["task_id"] = task_id
        op["params"] = json.dumps(params)
        if outbound_url and op.get("use_gradient_house"):
            op["outbound_service"] = outbound_url
    return cfg
