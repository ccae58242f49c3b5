"""Task configuration schema.

This is the task/config JSON API of the reference, reproduced field for
field: the canonical JSON <-> config converters in
ols_core/taskMgr/utils/utils.py:831-1027 (json2taskconfig) and
:1029-1197 (taskconfig2json).  A task JSON looks like:

{
  "user_id": "...", "task_id": "...",
  "target": {
    "priority": 0..10,
    "data": [{
      "name": "...", "data_path": "...", "data_split_type": bool,
      "data_transfer_type": "S3"|"MINIO"|"HTTP"|"FILE",
      "task_type": "...",
      "total_simulation": {"devices": [...], "nums": [...], "dynamic_nums": [...]},
      "allocation": {"optimization": bool,
                     "logical_simulation": [...], "device_simulation": [...],
                     "running_response": {"devices": [...], "nums": [...]}}
    }]},
  "operatorflow": {
    "flow_setting": {"round": R,
      "start": {"logical_simulation": {"strategy","wait_interval","total_timeout"},
                "device_simulation":  {...}},
      "stop":  {...}},
    "operators": [{
      "name": "...",
      "operation_behavior_controller": {"use_gradient_house": bool,
        "strategy_gradient_house": "...", "outbound_service": "..."},
      "input": [...], "use_data": bool,
      "model": {"use_model": bool, "model_for_train": bool,
                "model_transfer_type": "...", "model_path": "...",
                "model_update_style": "..."},
      "logical_simulation": {"operator_transfer_type", "operator_code_path",
                             "operator_entry_file", "operator_params"},
      "device_simulation": {...}}]},
  "logical_simulation": {
    "computation_unit": {"devices": [...], "setting": [{"num_cpus": n}]},
    "resource_request": [{"name","devices","num_request"}]},
  "device_simulation": {"resource_request": [...]}
}

Here the config objects are plain dataclasses instead of protobufs; the
converters keep the same key names, defaults and list semantics so tasks
written for the reference parse unchanged.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Dict, List

# Reference enum FileTransferType (ols_core/proto/taskService.proto).
FILE_TRANSFER_TYPES = ("S3", "HTTP", "FILE", "MINIO")


@dataclass
class TotalSimulation:
    devices: List[str] = field(default_factory=list)
    nums: List[int] = field(default_factory=list)
    dynamic_nums: List[int] = field(default_factory=list)


@dataclass
class RunningResponse:
    devices: List[str] = field(default_factory=list)
    nums: List[int] = field(default_factory=list)


@dataclass
class Allocation:
    optimization: bool = False
    logical_simulation: List[int] = field(default_factory=list)
    device_simulation: List[int] = field(default_factory=list)
    running_response: RunningResponse = field(default_factory=RunningResponse)


@dataclass
class TargetData:
    name: str = ""
    data_path: str = ""
    data_split_type: bool = False
    data_transfer_type: str = "S3"
    task_type: str = ""
    total_simulation: TotalSimulation = field(default_factory=TotalSimulation)
    allocation: Allocation = field(default_factory=Allocation)


@dataclass
class Target:
    data: List[TargetData] = field(default_factory=list)
    priority: int = 0


@dataclass
class StrategyCondition:
    strategy: str = ""
    wait_interval: int = 0
    total_timeout: int = 0


@dataclass
class FlowCondition:
    logical_simulation: StrategyCondition = field(default_factory=StrategyCondition)
    device_simulation: StrategyCondition = field(default_factory=StrategyCondition)


@dataclass
class FlowSetting:
    round: int = 0
    start: FlowCondition = field(default_factory=FlowCondition)
    stop: FlowCondition = field(default_factory=FlowCondition)


@dataclass
class BehaviorController:
    use_gradient_house: bool = False
    strategy_gradient_house: str = ""
    outbound_service: str = ""


@dataclass
class ModelInfo:
    use_model: bool = False
    model_for_train: bool = False
    model_transfer_type: str = "S3"
    model_path: str = ""
    model_update_style: str = ""


@dataclass
class OperatorSimulationInfo:
    operator_transfer_type: str = "S3"
    operator_code_path: str = ""
    operator_entry_file: str = ""
    operator_params: str = ""


@dataclass
class Operator:
    name: str = ""
    operation_behavior_controller: BehaviorController = field(default_factory=BehaviorController)
    input: List[str] = field(default_factory=list)
    use_data: bool = False
    model: ModelInfo = field(default_factory=ModelInfo)
    logical_simulation: OperatorSimulationInfo = field(default_factory=OperatorSimulationInfo)
    device_simulation: OperatorSimulationInfo = field(default_factory=OperatorSimulationInfo)


@dataclass
class OperatorFlow:
    flow_setting: FlowSetting = field(default_factory=FlowSetting)
    operators: List[Operator] = field(default_factory=list)


@dataclass
class UnitSetting:
    num_cpus: int = 0


@dataclass
class ComputationUnit:
    devices: List[str] = field(default_factory=list)
    setting: List[UnitSetting] = field(default_factory=list)


@dataclass
class ResourceRequest:
    name: str = ""
    devices: List[str] = field(default_factory=list)
    num_request: List[int] = field(default_factory=list)


@dataclass
class LogicalSimulation:
    computation_unit: ComputationUnit = field(default_factory=ComputationUnit)
    resource_request: List[ResourceRequest] = field(default_factory=list)


@dataclass
class DeviceSimulation:
    resource_request: List[ResourceRequest] = field(default_factory=list)


@dataclass
class TaskConfig:
    user_id: str = ""
    task_id: str = ""
    target: Target = field(default_factory=Target)
    operatorflow: OperatorFlow = field(default_factory=OperatorFlow)
    logical_simulation: LogicalSimulation = field(default_factory=LogicalSimulation)
    device_simulation: DeviceSimulation = field(default_factory=DeviceSimulation)


def _strategy_condition(d: Dict[str, Any]) -> StrategyCondition:
    return StrategyCondition(
        strategy=d.get("strategy", ""),
        wait_interval=d.get("wait_interval", 0),
        total_timeout=d.get("total_timeout", 0))


def json2taskconfig(jsonstring: str) -> TaskConfig:
    """Parse the task JSON (same keys/defaults as utils.py:831-1027)."""
    jd = json.loads(jsonstring)

    target_json = jd.get("target", {})
    data_list = []
    for data_index, dj in enumerate(target_json.get("data", [])):
        ts = dj.get("total_simulation", {})
        al = dj.get("allocation", {})
        rr = al.get("running_response", {})
        data_list.append(TargetData(
            name=dj.get("name", f"data_{data_index}"),
            data_path=dj.get("data_path", ""),
            data_split_type=dj.get("data_split_type", False),
            data_transfer_type=dj.get("data_transfer_type", "S3"),
            task_type=dj.get("task_type", ""),
            total_simulation=TotalSimulation(
                devices=list(ts.get("devices", [])),
                nums=list(ts.get("nums", [])),
                dynamic_nums=list(ts.get("dynamic_nums", []))),
            allocation=Allocation(
                optimization=al.get("optimization", False),
                logical_simulation=list(al.get("logical_simulation", [])),
                device_simulation=list(al.get("device_simulation", [])),
                running_response=RunningResponse(
                    devices=list(rr.get("devices", [])),
                    nums=list(rr.get("nums", []))))))
    target = Target(data=data_list, priority=target_json.get("priority", 0))

    of_json = jd.get("operatorflow", {})
    fs_json = of_json.get("flow_setting", {})
    start_json = fs_json.get("start", {})
    stop_json = fs_json.get("stop", {})
    flow_setting = FlowSetting(
        round=fs_json.get("round", 0),
        start=FlowCondition(
            logical_simulation=_strategy_condition(start_json.get("logical_simulation", {})),
            device_simulation=_strategy_condition(start_json.get("device_simulation", {}))),
        stop=FlowCondition(
            logical_simulation=_strategy_condition(stop_json.get("logical_simulation", {})),
            device_simulation=_strategy_condition(stop_json.get("device_simulation", {}))))

    operators = []
    for oj in of_json.get("operators", []):
        bc = oj.get("operation_behavior_controller", {})
        mj = oj.get("model", {})
        ls = oj.get("logical_simulation", {})
        ds = oj.get("device_simulation", {})
        op_input = oj.get("input", [])
        if op_input == "":
            op_input = []
        operators.append(Operator(
            name=oj.get("name", ""),
            operation_behavior_controller=BehaviorController(
                use_gradient_house=bc.get("use_gradient_house", False),
                strategy_gradient_house=bc.get("strategy_gradient_house", ""),
                outbound_service=bc.get("outbound_service", "")),
            input=list(op_input),
            use_data=oj.get("use_data", False),
            model=ModelInfo(
                use_model=mj.get("use_model", False),
                model_for_train=mj.get("model_for_train", False),
                model_transfer_type=mj.get("model_transfer_type", "S3"),
                model_path=mj.get("model_path", ""),
                model_update_style=mj.get("model_update_style", "")),
            logical_simulation=OperatorSimulationInfo(
                operator_transfer_type=ls.get("operator_transfer_type", "S3"),
                operator_code_path=ls.get("operator_code_path", ""),
                operator_entry_file=ls.get("operator_entry_file", ""),
                operator_params=ls.get("operator_params", "")),
            device_simulation=OperatorSimulationInfo(
                operator_transfer_type=ds.get("operator_transfer_type", "S3"),
                operator_code_path=ds.get("operator_code_path", ""),
                operator_entry_file=ds.get("operator_entry_file", ""),
                operator_params=ds.get("operator_params", ""))))

    ls_json = jd.get("logical_simulation", {})
    cu_json = ls_json.get("computation_unit", {})
    computation_unit = ComputationUnit(
        devices=list(cu_json.get("devices", [])),
        setting=[UnitSetting(num_cpus=s.get("num_cpus", 0))
                 for s in cu_json.get("setting", [])])
    logical_rr = [ResourceRequest(
        name=r.get("name", ""), devices=list(r.get("devices", [])),
        num_request=list(r.get("num_request", [])))
        for r in ls_json.get("resource_request", [])]

    ds_json = jd.get("device_simulation", {})
    device_rr = [ResourceRequest(
        name=r.get("name", ""), devices=list(r.get("devices", [])),
        num_request=list(r.get("num_request", [])))
        for r in ds_json.get("resource_request", [])]

    return TaskConfig(
        user_id=jd.get("user_id", ""),
        task_id=jd.get("task_id", ""),
        target=target,
        operatorflow=OperatorFlow(flow_setting=flow_setting, operators=operators),
        logical_simulation=LogicalSimulation(
            computation_unit=computation_unit, resource_request=logical_rr),
        device_simulation=DeviceSimulation(resource_request=device_rr))


def taskconfig2json(cfg: TaskConfig) -> str:
    """Serialize back to the canonical task JSON (utils.py:1029-1197)."""
    jd: Dict[str, Any] = {
        "user_id": cfg.user_id,
        "task_id": cfg.task_id,
        "target": {
            "priority": cfg.target.priority,
            "data": [{
                "name": d.name,
                "data_path": d.data_path,
                "data_split_type": d.data_split_type,
                "data_transfer_type": d.data_transfer_type,
                "task_type": d.task_type,
                "total_simulation": {
                    "devices": list(d.total_simulation.devices),
                    "nums": list(d.total_simulation.nums),
                    "dynamic_nums": list(d.total_simulation.dynamic_nums)},
                "allocation": {
                    "optimization": d.allocation.optimization,
                    "logical_simulation": list(d.allocation.logical_simulation),
                    "device_simulation": list(d.allocation.device_simulation),
                    "running_response": {
                        "devices": list(d.allocation.running_response.devices),
                        "nums": list(d.allocation.running_response.nums)}},
            } for d in cfg.target.data]},
        "operatorflow": {
            "flow_setting": {
                "round": cfg.operatorflow.flow_setting.round,
                "start": _cond_json(cfg.operatorflow.flow_setting.start),
                "stop": _cond_json(cfg.operatorflow.flow_setting.stop)},
            "operators": [{
                "name": op.name,
                "operation_behavior_controller": {
                    "use_gradient_house": op.operation_behavior_controller.use_gradient_house,
                    "strategy_gradient_house": op.operation_behavior_controller.strategy_gradient_house,
                    "outbound_service": op.operation_behavior_controller.outbound_service},
                "input": list(op.input),
                "use_data": op.use_data,
                "model": {
                    "use_model": op.model.use_model,
                    "model_for_train": op.model.model_for_train,
                    "model_transfer_type": op.model.model_transfer_type,
                    "model_path": op.model.model_path,
                    "model_update_style": op.model.model_update_style},
                "logical_simulation": _siminfo_json(op.logical_simulation),
                "device_simulation": _siminfo_json(op.device_simulation),
            } for op in cfg.operatorflow.operators]},
        "logical_simulation": {
            "computation_unit": {
                "devices": list(cfg.logical_simulation.computation_unit.devices),
                "setting": [{"num_cpus": s.num_cpus}
                            for s in cfg.logical_simulation.computation_unit.setting]},
            "resource_request": [_rr_json(r) for r in cfg.logical_simulation.resource_request]},
        "device_simulation": {
            "resource_request": [_rr_json(r) for r in cfg.device_simulation.resource_request]},
    }
    return json.dumps(jd, ensure_ascii=False)


def _cond_json(c: FlowCondition) -> Dict[str, Any]:
    return {
        "logical_simulation": {
            "strategy": c.logical_simulation.strategy,
            "wait_interval": c.logical_simulation.wait_interval,
            "total_timeout": c.logical_simulation.total_timeout},
        "device_simulation": {
            "strategy": c.device_simulation.strategy,
            "wait_interval": c.device_simulation.wait_interval,
            "total_timeout": c.device_simulation.total_timeout}}


def _siminfo_json(i: OperatorSimulationInfo) -> Dict[str, Any]:
    return {
        "operator_transfer_type": i.operator_transfer_type,
        "operator_code_path": i.operator_code_path,
        "operator_entry_file": i.operator_entry_file,
        "operator_params": i.operator_params}


def _rr_json(r: ResourceRequest) -> Dict[str, Any]:
    return {"name": r.name, "devices": list(r.devices),
            "num_request": list(r.num_request)}
