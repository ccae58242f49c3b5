"""Operator base classes: the operator-facing contract.

Parity with the reference's OperatorABC / OperatorRunScriptABC
(ols_core/taskMgr/base/base_operator.py:7-135,
base_operator_run_script.py:5-32): a custom operator receives a
``--params`` JSON with the documented schema (base_operator.py:12-53 —
task_id, config paths, current_round, data{name, data_path,
data_split_type, task_type, dataconfig}, operator{name,
operation_behavior_controller, input, use_data, model, operator_params},
actor_save_dir, actor_simulation_num, params) and implements
construct_run_params / construct_run_script / run.

Here an operator can also run IN PROCESS against the engine (no
subprocess per virtual device): subclass EngineOperator and override
``run_round`` — it is invoked once per operator-flow round with the
engine handle, replacing the reference's os.system per phone
(utils_run_task.py:496-514).
"""

from __future__ import annotations

import argparse
import json
from abc import ABC, abstractmethod
from typing import Any, Dict, List, Optional


class OperatorABC(ABC):
    """Script-style operator (reference base_operator.OperatorABC)."""

    def __init__(self):
        self.params: Dict[str, Any] = {}

    def get_params(self, argv: Optional[List[str]] = None) -> Dict[str, Any]:
        """Parse --params '<json>' (reference get_params)."""
        ap = argparse.ArgumentParser()
        ap.add_argument("--params", type=str, required=True)
        args, _ = ap.parse_known_args(argv)
        self.params = json.loads(args.params)
        return self.params

    @abstractmethod
    def construct_run_params(self) -> Dict[str, Any]:
        ...

    @abstractmethod
    def construct_run_script(self) -> str:
        ...

    @abstractmethod
    def run(self) -> int:
        ...


class OperatorRunScriptABC(ABC):
    """The entry-file side (reference base_operator_run_script)."""

    def __init__(self):
        self.params: Dict[str, Any] = {}

    def get_params(self, argv: Optional[List[str]] = None) -> Dict[str, Any]:
        ap = argparse.ArgumentParser()
        ap.add_argument("--params", type=str, required=True)
        args, _ = ap.parse_known_args(argv)
        self.params = json.loads(args.params)
        return self.params

    @abstractmethod
    def run(self) -> int:
        ...


class EngineOperator(ABC):
    """In-process operator: called once per round by the round loop."""

    name = "operator"

    @abstractmethod
    def run_round(self, engine, round_idx: int) -> Dict[str, Any]:
        """Execute this operator's work for one round; returns the
        per-round record (success/failed counts at minimum)."""
        ...
