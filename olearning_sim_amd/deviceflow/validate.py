"""Gradient-house strategy validation.

Parity with the reference's deviceflow ValidateParameters
(ols_core/deviceflow/utils/validate_parameters.py:24-233):

- exactly one of real_time_dispatch / flow_dispatch uses a strategy;
- flow_dispatch: total_dispatch_amount > 0; exactly one of
  specific_timing / specific_interval in use;
- specific_timing: timings/amounts same length; relative timings are
  non-negative and non-decreasing; absolute timings parse as
  "%Y-%m-%d %H:%M:%S" (per round when nested);
- specific_interval: intervals/domains/functions same length; interval
  bounds increase monotonically and do not overlap; each function
  evaluates with t bound (the reference eval-checks them);
- drop_simulation: at most one of drop_probability / drop_amounts;
  probabilities within [0, 1]; amounts >= 0; lengths match the
  timing/interval lists.
"""

from __future__ import annotations

import json
from datetime import datetime
from typing import Any, Dict, Optional

from .strategy import _eval_rate

_FMT = "%Y-%m-%d %H:%M:%S"


class ValidateStrategy:
    def __init__(self):
        self.last_error: Optional[str] = None

    def check(self, strategy: str) -> bool:
        try:
            self._check(json.loads(strategy))
        except (AssertionError, Exception) as e:  # noqa: B902
            self.last_error = str(e)
            return False
        self.last_error = None
        return True

    def _check(self, s: Dict[str, Any]) -> None:
        rt = s.get("real_time_dispatch", {})
        fl = s.get("flow_dispatch", {})
        rt_on = bool(rt.get("use_strategy", False))
        fl_on = bool(fl.get("use_strategy", False))
        assert rt_on != fl_on, \
            "exactly one of real_time_dispatch/flow_dispatch must be in use"
        if rt_on:
            sizes = rt.get("dispatch_batch_sizes", [])
            assert all(isinstance(x, int) and x > 0 for x in sizes), \
                "dispatch_batch_sizes must be positive ints"
            self._check_drop(rt.get("drop_simulation", {}), None)
            return
        total = fl.get("total_dispatch_amount", 0)
        assert isinstance(total, int) and total > 0, \
            "total_dispatch_amount must be a positive int"
        st = fl.get("specific_timing", {})
        si = fl.get("specific_interval", {})
        st_on = bool(st.get("use", False))
        si_on = bool(si.get("use", False))
        assert st_on != si_on, \
            "exactly one of specific_timing/specific_interval must be in use"
        if st_on:
            self._check_timing(st)
        else:
            self._check_interval(si)

    def _check_timing(self, st: Dict[str, Any]) -> None:
        time_type = st.get("time_type", "relative")
        timings = st.get("timings", [])
        amounts = st.get("amounts", [])
        assert len(amounts) > 0, "timing amounts must be non-empty"
        assert all(isinstance(a, int) and a >= 0 for a in amounts), \
            "amounts must be non-negative ints"
        if time_type == "relative":
            assert len(timings) == len(amounts), \
                "timings and amounts must have the same length"
            assert all(isinstance(t, (int, float)) and t >= 0 for t in timings), \
                "relative timings must be non-negative"
            assert all(timings[i] <= timings[i + 1]
                       for i in range(len(timings) - 1)), \
                "relative timings must be non-decreasing"
        else:
            assert time_type == "absolute", f"unknown time_type {time_type!r}"
            self._check_time_zone(st)
            for round_list in timings:
                assert len(round_list) == len(amounts), \
                    "each round's timings must match amounts length"
                for t in round_list:
                    datetime.strptime(t, _FMT)
        self._check_drop(st.get("drop_simulation", {}), len(amounts))

    @staticmethod
    def _check_time_zone(spec: Dict[str, Any]) -> None:
        tz = spec.get("time_zone")
        if tz:
            from zoneinfo import ZoneInfo
            try:
                ZoneInfo(tz)
            except Exception:
                raise AssertionError(f"unknown time_zone {tz!r}")

    def _check_interval(self, si: Dict[str, Any]) -> None:
        time_type = si.get("time_type", "relative")
        intervals = si.get("intervals", [])
        rules = si.get("dispatch_rules", {})
        domains = rules.get("domains", [])
        functions = rules.get("functions", [])
        assert len(domains) == len(functions) > 0, \
            "domains and functions must be non-empty and the same length"

        def check_rel(iv) -> None:
            assert len(iv) == len(domains), \
                "intervals must match domains length"
            prev_end = None
            for lo, hi in iv:
                assert isinstance(lo, int) and isinstance(hi, int) and lo < hi, \
                    f"interval [{lo},{hi}] must be increasing ints"
                assert lo >= 0, "intervals must be non-negative"
                if prev_end is not None:
                    assert lo >= prev_end, "intervals must not overlap"
                prev_end = hi

        if time_type == "relative":
            check_rel(intervals)
        else:
            assert time_type == "absolute", f"unknown time_type {time_type!r}"
            self._check_time_zone(si)
            for round_iv in intervals:
                prev_end = None
                assert len(round_iv) == len(domains)
                for lo, hi in round_iv:
                    t0 = datetime.strptime(lo, _FMT)
                    t1 = datetime.strptime(hi, _FMT)
                    assert t0 < t1, "absolute interval must increase"
                    if prev_end is not None:
                        assert t0 >= prev_end, "intervals must not overlap"
                    prev_end = t1
        for lo, hi in domains:
            assert float(lo) <= float(hi), "domain bounds must be ordered"
        for f in functions:
            _eval_rate(f, 0.5)   # must evaluate with t bound
        self._check_drop(si.get("drop_simulation", {}), len(domains))

    @staticmethod
    def _check_drop(drop: Dict[str, Any], expect_len) -> None:
        if not drop:
            return
        assert len(drop) == 1, \
            "drop_simulation allows only one of drop_probability/drop_amounts"
        if "drop_probability" in drop:
            probs = drop["drop_probability"]
            if isinstance(probs, list):
                assert expect_len is None or len(probs) == expect_len, \
                    "drop_probability length mismatch"
                vals = probs
            else:
                vals = [probs]
            assert all(0 <= p <= 1 for p in vals), \
                "drop_probability must be within [0, 1]"
        elif "drop_amounts" in drop:
            amts = drop["drop_amounts"]
            assert expect_len is None or len(amts) == expect_len, \
                "drop_amounts length mismatch"
            assert all(isinstance(a, int) and a >= 0 for a in amts), \
                "drop_amounts must be non-negative ints"
        else:
            raise AssertionError("unknown drop_simulation key")
