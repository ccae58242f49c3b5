"""Script-file operator execution.

Parity with the reference's Actor subprocess path
(ols_core/taskMgr/utils/utils_run_task.py:481-514 `loop_run` /
`_single_compute_step`): user operator code staged into the task
working directory is executed as `python3 <entry> --params '<json>'`
with the documented params schema (base_operator.py:12-53).  The
reference forks once per virtual phone; here one invocation simulates a
contiguous SHARD of virtual devices (`actor_simulation_num` tells the
script how many), shards run concurrently, and a shard reports
fine-grained counts by writing `result.json` in its save dir —
exit code 0 without a result file means every device in the shard
succeeded, a non-zero exit fails the whole shard.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Dict, List, Optional, Tuple


class ScriptOperator:
    def __init__(self, name: str, staged_dir: str, entry_file: str,
                 operator_params: str, task_id: str, work_dir: str,
                 clients: int, shards: int = 4,
                 data_info: Optional[Dict[str, Any]] = None,
                 model_info: Optional[Dict[str, Any]] = None,
                 timeout: float = 300.0, max_workers: int = 8):
        self.name = name
        self.staged_dir = staged_dir
        self.entry_file = entry_file
        self.operator_params = operator_params
        self.task_id = task_id
        self.work_dir = work_dir
        self.clients = max(1, clients)
        self.shards = max(1, min(shards, self.clients))
        self.data_info = data_info or {}
        self.model_info = model_info or {}
        self.timeout = timeout
        self.max_workers = max_workers

    # -- sharding ---------------------------------------------------------
    def shard_ranges(self) -> List[Tuple[int, int]]:
        """Contiguous client ranges, one per invocation (the analogue of
        construct_run_params' split_index_list, run_task.py:62-106)."""
        base, rem = divmod(self.clients, self.shards)
        ranges, lo = [], 0
        for s in range(self.shards):
            hi = lo + base + (1 if s < rem else 0)
            ranges.append((lo, hi))
            lo = hi
        return ranges

    def _params_for(self, round_idx: int, shard: int,
                    lo: int, hi: int,
                    model_path: Optional[str] = None) -> Dict[str, Any]:
        save_dir = os.path.join(self.work_dir, f"round_{round_idx}",
                                f"shard_{shard}")
        os.makedirs(save_dir, exist_ok=True)
        model = dict(self.model_info)
        if model_path:
            # round r>0 weights under the templated model_update_style
            # name (reference download_model_files,
            # utils_run_task.py:327-397)
            model["current_model_path"] = model_path
        return {
            "task_id": self.task_id,
            "current_round": round_idx,
            "data": dict(self.data_info),
            "operator": {
                "name": self.name,
                "use_data": bool(self.data_info),
                "model": model,
                "operator_params": self.operator_params,
            },
            "actor_save_dir": save_dir,
            "actor_simulation_num": hi - lo,
            "client_range": [lo, hi],
            "params": self.operator_params,
        }

    # -- execution --------------------------------------------------------
    def _run_shard(self, round_idx: int, shard: int, lo: int, hi: int,
                   model_path: Optional[str] = None) -> Tuple[int, int, int]:
        """Returns (lo, hi, success_count)."""
        params = self._params_for(round_idx, shard, lo, hi, model_path)
        cmd = [sys.executable, os.path.join(self.staged_dir, self.entry_file),
               "--params", json.dumps(params)]
        try:
            proc = subprocess.run(cmd, cwd=self.staged_dir,
                                  capture_output=True, timeout=self.timeout)
            rc = proc.returncode
        except subprocess.TimeoutExpired:
            rc = -1
        result_file = os.path.join(params["actor_save_dir"], "result.json")
        if os.path.exists(result_file):
            try:
                with open(result_file) as f:
                    res = json.load(f)
                succ = int(res.get("success", hi - lo))
                return lo, hi, max(0, min(hi - lo, succ))
            except (ValueError, OSError):
                pass
        return lo, hi, (hi - lo) if rc == 0 else 0

    def run_round(self, round_idx: int,
                  model_path: Optional[str] = None) -> Dict[str, Any]:
        """Run every shard; failed devices are attributed to the TAIL of
        their shard's client range so per-tier accounting stays exact."""
        ranges = self.shard_ranges()
        with ThreadPoolExecutor(
                max_workers=min(self.max_workers, len(ranges))) as pool:
            results = list(pool.map(
                lambda args: self._run_shard(round_idx, *args, model_path),
                [(s, lo, hi) for s, (lo, hi) in enumerate(ranges)]))
        success = sum(r[2] for r in results)
        failed = self.clients - success
        # failed client-id ranges (tail of each shard)
        failed_ranges = [(lo + s, hi) for lo, hi, s in results if lo + s < hi]
        return {"success": success, "failed": failed,
                "failed_ranges": failed_ranges}


def per_tier_counts(failed_ranges: List[Tuple[int, int]],
                    tier_bounds: List[int]) -> Tuple[List[int], List[int]]:
    """Intersect failed client-id ranges with tier prefix ranges to get
    (success_per_tier, failed_per_tier)."""
    T = len(tier_bounds) - 1
    fail_t = [0] * T
    for lo, hi in failed_ranges:
        for t in range(T):
            a, b = tier_bounds[t], tier_bounds[t + 1]
            fail_t[t] += max(0, min(hi, b) - max(lo, a))
    succ_t = [(tier_bounds[t + 1] - tier_bounds[t]) - fail_t[t]
              for t in range(T)]
    return succ_t, fail_t
