"""Three-stage task validation.

Same contract as the reference's ValidateParameters
(ols_core/taskMgr/utils/utils.py:283-811): a task is admitted only if it
passes (1) type checks, (2) value-correctness checks, (3) cross-field
relationship checks.  Each stage returns False (with the failure reason
recorded) rather than raising, as the gRPC servicer expects.

The rules preserved stage by stage:

type (utils.py:283-399)
  - every scalar field has the schema's type; list fields are lists of
    the right element type (str for device tiers, int for counts).
correctness (utils.py:401-554)
  - user_id/task_id non-empty, ASCII-printable;
  - data names non-empty, ASCII; data_path (when set) is .zip or a
    plain path matching [a-zA-Z0-9/._-]+; transfer types in enum;
  - total_simulation.devices non-empty, unique, ASCII; nums > 0;
    dynamic_nums >= 0; allocation lists >= 0; running_response devices
    unique/ASCII, nums >= 0; priority in [0,10];
  - flow_setting.round > 0, wait_interval/total_timeout >= 0;
  - operator names non-empty, ASCII, no spaces; operator params (when
    set) parse as JSON; logical operator entry file .py; code path a
    dir or .zip (device side .apk);
  - computation_unit devices unique/ASCII, num_cpus >= 1;
    resource_request names non-empty/ASCII, devices unique, nums >= 0.
relationship (utils.py:556-811)
  - len(devices)==len(nums)==len(dynamic_nums); nums > dynamic_nums
    elementwise; running_response ⊆ devices, rr nums <= nums;
  - when optimization==False: nums == logical_alloc + device_alloc and
    device_alloc >= running_response (elementwise);
  - wait_interval <= total_timeout for all four flow conditions;
  - gradient-house strategy non-empty when use_gradient_house;
  - operator.input names must appear among earlier operators;
  - model_path non-empty when use_model;
  - per operator, logical+device code path (and entry file) not both
    empty;
  - the set of resource_request data names (logical+device) equals the
    set of target data names;
  - computation_unit devices cover every total_simulation device and
    len(devices)==len(setting);
  - per-data per-device request counts consistent with allocation and
    running_response (utils.py:704-805 semantics).
"""

from __future__ import annotations

import json
import os
import re
from typing import Any, Dict, List, Optional

from .schema import TaskConfig, FILE_TRANSFER_TYPES

_PATH_RE = re.compile(r"^[a-zA-Z0-9/._-]+$")


def _ascii_ok(s: str) -> bool:
    return all(32 <= ord(ch) <= 126 for ch in s)


def _has_ext(s: str, ext: str) -> bool:
    return s.endswith(ext) and len(s) > len(ext)


def _path_safe(s: str) -> bool:
    """True when s can be joined into a work/checkpoint directory
    without escaping it (staging.py rmtree's work_dir/<name>;
    checkpoint.py joins the rendered model_update_style)."""
    return "/" not in s and "\\" not in s and ".." not in s


class ValidationError(Exception):
    pass


class ValidateParameters:
    """Validate a raw task-JSON dict and its parsed TaskConfig."""

    def __init__(self):
        self.last_error: Optional[str] = None

    # ------------------------------------------------------------------
    def validate_task_parameters(self, raw: Dict[str, Any],
                                 cfg: TaskConfig) -> bool:
        try:
            self.validate_type(raw)
            self._stage_correctness(cfg)
            self._stage_relationship(cfg)
        except (ValidationError, AssertionError) as e:
            self.last_error = str(e)
            return False
        self.last_error = None
        return True

    # -- stage 1: types --------------------------------------------------
    def validate_type(self, raw: Dict[str, Any]) -> None:
        def need(cond: bool, msg: str) -> None:
            if not cond:
                raise ValidationError(msg)

        need(isinstance(raw.get("user_id", ""), str), "user_id must be a string")
        need(isinstance(raw.get("task_id", ""), str), "task_id must be a string")
        target = raw.get("target", {})
        need(isinstance(target, dict), "target must be an object")
        need(isinstance(target.get("priority", 0), int), "target.priority must be an int")
        for i, d in enumerate(target.get("data", [])):
            where = f"target.data[{i}]"
            need(isinstance(d.get("name", ""), str), f"{where}.name must be a string")
            need(isinstance(d.get("data_path", ""), str), f"{where}.data_path must be a string")
            need(isinstance(d.get("data_split_type", False), bool),
                 f"{where}.data_split_type must be a bool")
            need(isinstance(d.get("data_transfer_type", "S3"), str),
                 f"{where}.data_transfer_type must be a string")
            need(isinstance(d.get("task_type", ""), str), f"{where}.task_type must be a string")
            ts = d.get("total_simulation", {})
            need(all(isinstance(x, str) for x in ts.get("devices", [])),
                 f"{where}.total_simulation.devices must be strings")
            need(all(isinstance(x, int) for x in ts.get("nums", [])),
                 f"{where}.total_simulation.nums must be ints")
            need(all(isinstance(x, int) for x in ts.get("dynamic_nums", [])),
                 f"{where}.total_simulation.dynamic_nums must be ints")
            al = d.get("allocation", {})
            need(isinstance(al.get("optimization", False), bool),
                 f"{where}.allocation.optimization must be a bool")
            need(all(isinstance(x, int) for x in al.get("logical_simulation", [])),
                 f"{where}.allocation.logical_simulation must be ints")
            need(all(isinstance(x, int) for x in al.get("device_simulation", [])),
                 f"{where}.allocation.device_simulation must be ints")
            rr = al.get("running_response", {})
            need(all(isinstance(x, str) for x in rr.get("devices", [])),
                 f"{where}.allocation.running_response.devices must be strings")
            need(all(isinstance(x, int) for x in rr.get("nums", [])),
                 f"{where}.allocation.running_response.nums must be ints")

        of = raw.get("operatorflow", {})
        fs = of.get("flow_setting", {})
        need(isinstance(fs.get("round", 0), int), "flow_setting.round must be an int")
        for phase in ("start", "stop"):
            for side in ("logical_simulation", "device_simulation"):
                c = fs.get(phase, {}).get(side, {})
                need(isinstance(c.get("strategy", ""), str),
                     f"flow_setting.{phase}.{side}.strategy must be a string")
                need(isinstance(c.get("wait_interval", 0), int),
                     f"flow_setting.{phase}.{side}.wait_interval must be an int")
                need(isinstance(c.get("total_timeout", 0), int),
                     f"flow_setting.{phase}.{side}.total_timeout must be an int")
        for i, op in enumerate(of.get("operators", [])):
            where = f"operators[{i}]"
            need(isinstance(op.get("name", ""), str), f"{where}.name must be a string")
            bc = op.get("operation_behavior_controller", {})
            need(isinstance(bc.get("use_gradient_house", False), bool),
                 f"{where}.use_gradient_house must be a bool")
            need(isinstance(bc.get("strategy_gradient_house", ""), str),
                 f"{where}.strategy_gradient_house must be a string")
            need(isinstance(bc.get("outbound_service", ""), str),
                 f"{where}.outbound_service must be a string")
            op_input = op.get("input", [])
            need(op_input == "" or (isinstance(op_input, list)
                                    and all(isinstance(x, str) for x in op_input)),
                 f"{where}.input must be a list of strings")
            need(isinstance(op.get("use_data", False), bool),
                 f"{where}.use_data must be a bool")
            m = op.get("model", {})
            need(isinstance(m.get("use_model", False), bool),
                 f"{where}.model.use_model must be a bool")
            need(isinstance(m.get("model_for_train", False), bool),
                 f"{where}.model.model_for_train must be a bool")
            need(isinstance(m.get("model_path", ""), str),
                 f"{where}.model.model_path must be a string")
            need(isinstance(m.get("model_update_style", ""), str),
                 f"{where}.model.model_update_style must be a string")
            for side in ("logical_simulation", "device_simulation"):
                si = op.get(side, {})
                for k in ("operator_transfer_type", "operator_code_path",
                          "operator_entry_file", "operator_params"):
                    need(isinstance(si.get(k, ""), str),
                         f"{where}.{side}.{k} must be a string")

        ls = raw.get("logical_simulation", {})
        cu = ls.get("computation_unit", {})
        need(all(isinstance(x, str) for x in cu.get("devices", [])),
             "computation_unit.devices must be strings")
        for s in cu.get("setting", []):
            need(isinstance(s.get("num_cpus", 0), int),
                 "computation_unit.setting.num_cpus must be an int")
        for sect_name, sect in (("logical_simulation", ls),
                                ("device_simulation", raw.get("device_simulation", {}))):
            for i, r in enumerate(sect.get("resource_request", [])):
                where = f"{sect_name}.resource_request[{i}]"
                need(isinstance(r.get("name", ""), str), f"{where}.name must be a string")
                need(all(isinstance(x, str) for x in r.get("devices", [])),
                     f"{where}.devices must be strings")
                need(all(isinstance(x, int) for x in r.get("num_request", [])),
                     f"{where}.num_request must be ints")

    # -- stage 2: value correctness --------------------------------------
    def _stage_correctness(self, cfg: TaskConfig) -> None:
        def need(cond: bool, msg: str) -> None:
            if not cond:
                raise ValidationError(msg)

        need(cfg.user_id != "", "user_id must not be empty")
        need(_ascii_ok(cfg.user_id), "user_id contains non-ASCII characters")
        need(cfg.task_id != "", "task_id must not be empty")
        need(_ascii_ok(cfg.task_id), "task_id contains non-ASCII characters")
        need(_path_safe(cfg.task_id) and cfg.task_id not in (".", ".."),
             "task_id must not contain path separators or '..'")

        for i, d in enumerate(cfg.target.data):
            where = f"target.data[{i}] ({d.name!r})"
            need(d.name != "", f"target.data[{i}].name must not be empty")
            need(_ascii_ok(d.name), f"{where}: name has non-ASCII characters")
            if d.data_path:
                need(_has_ext(d.data_path, ".zip") or bool(_PATH_RE.match(d.data_path)),
                     f"{where}: data_path must be a .zip file or a plain directory path")
            need(d.data_transfer_type in FILE_TRANSFER_TYPES,
                 f"{where}: unknown data_transfer_type {d.data_transfer_type!r}")
            need(_ascii_ok(d.task_type), f"{where}: task_type has non-ASCII characters")
            devs = d.total_simulation.devices
            need(len(devs) > 0, f"{where}: total_simulation.devices must be non-empty")
            need(len(devs) == len(set(devs)),
                 f"{where}: total_simulation.devices has duplicates")
            need(all(_ascii_ok(x) for x in devs),
                 f"{where}: device tiers have non-ASCII characters")
            need(all(x > 0 for x in d.total_simulation.nums),
                 f"{where}: total_simulation.nums must be > 0")
            need(all(x >= 0 for x in d.total_simulation.dynamic_nums),
                 f"{where}: dynamic_nums must be >= 0")
            need(all(x >= 0 for x in d.allocation.logical_simulation),
                 f"{where}: allocation.logical_simulation must be >= 0")
            need(all(x >= 0 for x in d.allocation.device_simulation),
                 f"{where}: allocation.device_simulation must be >= 0")
            rr = d.allocation.running_response
            need(all(_ascii_ok(x) for x in rr.devices),
                 f"{where}: running_response devices have non-ASCII characters")
            need(len(rr.devices) == len(set(rr.devices)),
                 f"{where}: running_response.devices has duplicates")
            need(all(x >= 0 for x in rr.nums),
                 f"{where}: running_response.nums must be >= 0")
        need(0 <= cfg.target.priority <= 10, "target.priority must be in [0, 10]")

        fs = cfg.operatorflow.flow_setting
        need(fs.round > 0, "flow_setting.round must be > 0")
        for phase_name, cond in (("start", fs.start), ("stop", fs.stop)):
            for side_name, sc in (("logical_simulation", cond.logical_simulation),
                                  ("device_simulation", cond.device_simulation)):
                where = f"flow_setting.{phase_name}.{side_name}"
                need(_ascii_ok(sc.strategy), f"{where}.strategy has non-ASCII characters")
                need(sc.wait_interval >= 0, f"{where}.wait_interval must be >= 0")
                need(sc.total_timeout >= 0, f"{where}.total_timeout must be >= 0")

        for i, op in enumerate(cfg.operatorflow.operators):
            where = f"operators[{i}] ({op.name!r})"
            need(op.name != "", f"operators[{i}].name must not be empty")
            need(_ascii_ok(op.name), f"{where}: name has non-ASCII characters")
            need(" " not in op.name, f"{where}: name must not contain spaces")
            need(_path_safe(op.name) and op.name not in (".", ".."),
                 f"{where}: name must not contain path separators or '..'")
            bc = op.operation_behavior_controller
            need(_ascii_ok(bc.strategy_gradient_house),
                 f"{where}: strategy_gradient_house has non-ASCII characters")
            need(_ascii_ok(bc.outbound_service),
                 f"{where}: outbound_service has non-ASCII characters")
            need(all(_ascii_ok(x) for x in op.input),
                 f"{where}: input entries have non-ASCII characters")
            need(op.model.model_transfer_type in FILE_TRANSFER_TYPES,
                 f"{where}: unknown model_transfer_type")
            need(_ascii_ok(op.model.model_path),
                 f"{where}: model_path has non-ASCII characters")
            need(_ascii_ok(op.model.model_update_style),
                 f"{where}: model_update_style has non-ASCII characters")
            need(_path_safe(op.model.model_update_style),
                 f"{where}: model_update_style must not contain path "
                 f"separators or '..'")
            lsim = op.logical_simulation
            need(lsim.operator_transfer_type in FILE_TRANSFER_TYPES,
                 f"{where}: unknown logical operator_transfer_type")
            if lsim.operator_code_path:
                need(_ascii_ok(lsim.operator_code_path),
                     f"{where}: logical operator_code_path has non-ASCII characters")
                need(os.path.isdir(os.path.abspath(lsim.operator_code_path))
                     or _has_ext(lsim.operator_code_path, ".zip")
                     or lsim.operator_code_path.startswith("builtin:"),
                     f"{where}: logical operator_code_path must be an existing "
                     f"directory, a .zip file, or a builtin: operator name")
            if lsim.operator_entry_file:
                need(_ascii_ok(lsim.operator_entry_file),
                     f"{where}: logical operator_entry_file has non-ASCII characters")
                need(_has_ext(lsim.operator_entry_file, ".py"),
                     f"{where}: logical operator_entry_file must end in .py")
            if lsim.operator_params:
                try:
                    json.loads(lsim.operator_params)
                except Exception:
                    raise ValidationError(
                        f"{where}: logical operator_params must be a JSON string")
            dsim = op.device_simulation
            need(dsim.operator_transfer_type in FILE_TRANSFER_TYPES,
                 f"{where}: unknown device operator_transfer_type")
            if dsim.operator_code_path:
                need(_ascii_ok(dsim.operator_code_path),
                     f"{where}: device operator_code_path has non-ASCII characters")
                need(_has_ext(dsim.operator_code_path, ".apk"),
                     f"{where}: device operator_code_path must end in .apk")
            if dsim.operator_entry_file:
                need(_ascii_ok(dsim.operator_entry_file),
                     f"{where}: device operator_entry_file has non-ASCII characters")
                need(_has_ext(dsim.operator_entry_file, ".apk"),
                     f"{where}: device operator_entry_file must end in .apk")
            if dsim.operator_params:
                try:
                    json.loads(dsim.operator_params)
                except Exception:
                    raise ValidationError(
                        f"{where}: device operator_params must be a JSON string")

        cu = cfg.logical_simulation.computation_unit
        need(len(cu.devices) == len(set(cu.devices)),
             "computation_unit.devices has duplicates")
        need(all(_ascii_ok(x) for x in cu.devices),
             "computation_unit.devices have non-ASCII characters")
        need(all(s.num_cpus >= 1 for s in cu.setting),
             "computation_unit.setting.num_cpus must be >= 1")
        for sect_name, rr_list in (
                ("logical_simulation", cfg.logical_simulation.resource_request),
                ("device_simulation", cfg.device_simulation.resource_request)):
            for i, r in enumerate(rr_list):
                where = f"{sect_name}.resource_request[{i}]"
                need(r.name != "", f"{where}.name must not be empty")
                need(_ascii_ok(r.name), f"{where}.name has non-ASCII characters")
                need(len(r.devices) == len(set(r.devices)),
                     f"{where}.devices has duplicates")
                need(all(_ascii_ok(x) for x in r.devices),
                     f"{where}.devices have non-ASCII characters")
                need(all(x >= 0 for x in r.num_request),
                     f"{where}.num_request must be >= 0")

    # -- stage 3: relationships ------------------------------------------
    def _stage_relationship(self, cfg: TaskConfig) -> None:
        def need(cond: bool, msg: str) -> None:
            if not cond:
                raise ValidationError(msg)

        data_names: List[str] = []
        for d in cfg.target.data:
            data_names.append(d.name)
            if d.data_path and d.data_transfer_type not in ("MINIO", "FILE"):
                need(_has_ext(d.data_path, ".zip"),
                     f"data {d.name!r}: non-MINIO/FILE data_path must be a .zip file")
            devs = d.total_simulation.devices
            nums = d.total_simulation.nums
            dnums = d.total_simulation.dynamic_nums
            need(len(devs) == len(nums) == len(dnums),
                 f"data {d.name!r}: devices/nums/dynamic_nums lengths differ")
            need(all(n > z for n, z in zip(nums, dnums)),
                 f"data {d.name!r}: each nums entry must exceed dynamic_nums")
            rr = d.allocation.running_response
            need(set(rr.devices).issubset(set(devs)),
                 f"data {d.name!r}: running_response devices not all in total_simulation")
            need(len(rr.devices) == len(rr.nums),
                 f"data {d.name!r}: running_response devices/nums lengths differ")
            rr_map = dict(zip(rr.devices, rr.nums))
            rr_reorder = [rr_map.get(dev, 0) for dev in devs]
            need(all(r <= n for r, n in zip(rr_reorder, nums)),
                 f"data {d.name!r}: running_response nums exceed total nums")
            if not d.allocation.optimization:
                la = list(d.allocation.logical_simulation) or [0] * len(nums)
                da = list(d.allocation.device_simulation) or [0] * len(nums)
                need(len(la) == len(nums) == len(da),
                     f"data {d.name!r}: allocation list lengths differ from nums")
                need(all(n == x + y for n, x, y in zip(nums, la, da)),
                     f"data {d.name!r}: logical+device allocation must equal nums")
                need(all(y >= r for y, r in zip(da, rr_reorder)),
                     f"data {d.name!r}: device allocation below running_response")

        fs = cfg.operatorflow.flow_setting
        for cond in (fs.start.logical_simulation, fs.start.device_simulation,
                     fs.stop.logical_simulation, fs.stop.device_simulation):
            need(cond.wait_interval <= cond.total_timeout,
                 "flow_setting: wait_interval must not exceed total_timeout")

        seen_ops: List[str] = []
        for op in cfg.operatorflow.operators:
            if op.operation_behavior_controller.use_gradient_house:
                need(op.operation_behavior_controller.strategy_gradient_house != "",
                     f"operator {op.name!r}: gradient-house strategy required "
                     f"when use_gradient_house is set")
            if op.input:
                need(set(op.input).issubset(set(seen_ops)),
                     f"operator {op.name!r}: input refers to operators not yet defined")
            if op.model.use_model:
                need(op.model.model_path != "",
                     f"operator {op.name!r}: model_path required when use_model is set")
            lpath = op.logical_simulation.operator_code_path
            if lpath and os.path.isdir(os.path.abspath(lpath)):
                need(op.logical_simulation.operator_transfer_type == "FILE",
                     f"operator {op.name!r}: directory code path requires FILE transfer")
            need(not (op.logical_simulation.operator_code_path == ""
                      and op.device_simulation.operator_code_path == ""),
                 f"operator {op.name!r}: logical and device code paths both empty")
            need(not (op.logical_simulation.operator_entry_file == ""
                      and op.device_simulation.operator_entry_file == ""),
                 f"operator {op.name!r}: logical and device entry files both empty")
            seen_ops.append(op.name)

        rr_names = ([r.name for r in cfg.logical_simulation.resource_request]
                    + [r.name for r in cfg.device_simulation.resource_request])
        need(set(rr_names) == set(data_names),
             "resource_request data names must exactly cover the target data names")

        cu = cfg.logical_simulation.computation_unit
        need(len(cu.devices) == len(cu.setting),
             "computation_unit devices/setting lengths differ")
        all_devices = [x for d in cfg.target.data for x in d.total_simulation.devices]
        need(set(all_devices).issubset(set(cu.devices)),
             "computation_unit.devices must cover every total_simulation device")

        # per-data per-device request consistency (utils.py:704-805)
        by_name = {d.name: d for d in cfg.target.data}
        for r in cfg.logical_simulation.resource_request:
            need(r.name in by_name,
                 f"logical resource_request {r.name!r} not in target data")
            need(len(r.devices) == len(r.num_request),
                 f"logical resource_request {r.name!r}: devices/nums lengths differ")
            d = by_name[r.name]
            req_map = dict(zip(r.devices, r.num_request))
            if not d.allocation.optimization:
                alloc_map = dict(zip(d.total_simulation.devices,
                                     d.allocation.logical_simulation
                                     or [0] * len(d.total_simulation.devices)))
            else:
                alloc_map = {}
            for dev, n_req in req_map.items():
                n_alloc = alloc_map.get(dev, 0)
                if not d.allocation.optimization and n_alloc > 0:
                    need(n_req > 0,
                         f"data {r.name!r}: logical request for {dev!r} must be > 0 "
                         f"when that tier has allocated machine-times")
                else:
                    need(n_req >= 0,
                         f"data {r.name!r}: logical request for {dev!r} must be >= 0")

        for r in cfg.device_simulation.resource_request:
            need(r.name in by_name,
                 f"device resource_request {r.name!r} not in target data")
            need(len(r.devices) == len(r.num_request),
                 f"device resource_request {r.name!r}: devices/nums lengths differ")
            d = by_name[r.name]
            req_map = dict(zip(r.devices, r.num_request))
            rr_map = dict(zip(d.allocation.running_response.devices,
                              d.allocation.running_response.nums))
            if not d.allocation.optimization:
                alloc_map = dict(zip(d.total_simulation.devices,
                                     d.allocation.device_simulation
                                     or [0] * len(d.total_simulation.devices)))
                for dev, n_alloc in alloc_map.items():
                    n_req = req_map.get(dev, 0)
                    n_rr = rr_map.get(dev, 0)
                    if n_alloc == n_rr:
                        need(n_req >= n_rr,
                             f"data {r.name!r}: device request for {dev!r} below "
                             f"running_response")
                    else:
                        need(n_req >= 1,
                             f"data {r.name!r}: device request for {dev!r} must be >= 1")
                        need(n_req > n_rr,
                             f"data {r.name!r}: device request for {dev!r} must exceed "
                             f"running_response")
            else:
                for dev, n_req in req_map.items():
                    n_rr = rr_map.get(dev, 0)
                    if n_rr > 0:
                        need(n_req > n_rr,
                             f"data {r.name!r}: device request for {dev!r} must exceed "
                             f"running_response under optimization")
