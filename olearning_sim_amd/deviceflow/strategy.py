"""Deviceflow behaviour-model math (the gradient-house strategy).

Functional parity with the reference's Strategy
(ols_core/deviceflow/non_grpc/strategy.py:19-445), the core of its
device-behaviour simulation.  A strategy JSON has exactly one of:

  {"real_time_dispatch": {"use_strategy": true,
      "dispatch_batch_sizes": [..],
      "drop_simulation": {"drop_probability": p}}}

  {"flow_dispatch": {"use_strategy": true,
      "total_dispatch_amount": N,
      "specific_timing": {"use": true, "time_type": "relative"|"absolute",
          "time_zone": "...", "timings": [...], "amounts": [...],
          "drop_simulation": {...}},
      "specific_interval": {"use": true, "time_type": ...,
          "intervals": [[t0,t1],...],
          "dispatch_rules": {"domains": [[d0,d1],...],
                             "functions": ["math.sin(t)+1", ...]},
          "drop_simulation": {"drop_probability": [..] | "drop_amounts": [..]}}}}

flow analysis returns (dispatch_timing, dispatch_amount,
drop_simulation_list): per-send-slot waits (seconds, successive deltas),
per-slot send counts, and per-slot sorted lists of in-slot message
indices to drop.  The interval form integrates each arrival-rate
function f(t) over its domain by trapezoids at 100 sub-steps per
1-second slot (only positive area counts), apportions
total_dispatch_amount across intervals by area share (round-half-even,
remainder to the last interval), then splits each interval's amount over
its 1-second slots by area with a residual-carry rounding — identical
slot counts to the reference for identical inputs (verified by the
tests in tests/test_strategy_math.py).
"""

from __future__ import annotations

import json
import math
import random as _random
from datetime import datetime
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

AREA_CALCULATION_NUM = 100  # integration sub-steps per 1-second slot

_SAFE_EVAL_GLOBALS = {"math": math, "np": np, "abs": abs, "min": min,
                      "max": max, "pow": pow, "__builtins__": {}}

DEFAULT_TIME_ZONE = "Asia/Shanghai"   # reference strategy.py:116-121


def _zone_now(spec: Dict[str, Any]) -> datetime:
    """Wall-clock 'now' in the strategy's configured time zone (naive,
    for comparison against the spec's absolute timestamps).  The
    reference converts via pytz with Asia/Shanghai as the default; a
    host whose local TZ differs would otherwise shift every absolute
    schedule by the TZ offset."""
    tz_name = spec.get("time_zone") or DEFAULT_TIME_ZONE
    try:
        from zoneinfo import ZoneInfo
        return datetime.now(ZoneInfo(tz_name)).replace(tzinfo=None)
    except Exception:
        return datetime.now()


# -- arrival-rate expression compiler ----------------------------------
# Rate functions are user task input ("math.sin(t)+1").  Instead of a
# scrubbed-globals eval (escapable via attribute chains), parse the
# expression and allow only arithmetic, numeric literals, the variable
# t, calls on math.*/np.* and abs/min/max/pow — then compile the vetted
# AST once and cache the code object (the integrator calls f(t) 100x
# per schedule slot).

_ALLOWED_NODES = (
    "Expression", "BinOp", "UnaryOp", "Call", "Constant", "Name", "Load",
    "Attribute", "Add", "Sub", "Mult", "Div", "FloorDiv", "Mod", "Pow",
    "USub", "UAdd", "IfExp", "Compare", "Lt", "LtE", "Gt", "GtE", "Eq",
    "NotEq", "BoolOp", "And", "Or", "Tuple",
)
_ALLOWED_NAMES = {"t", "abs", "min", "max", "pow", "math", "np"}
_EXPR_CACHE: Dict[str, Any] = {}


class RateExprError(ValueError):
    pass


def compile_rate_expr(func_string: str):
    """Compile an arrival-rate expression to a code object, allowing
    only whitelisted AST nodes/names.  Raises RateExprError otherwise."""
    code = _EXPR_CACHE.get(func_string)
    if code is not None:
        return code
    import ast
    try:
        tree = ast.parse(func_string, mode="eval")
    except SyntaxError as e:
        raise RateExprError(f"invalid rate expression: {e}") from None
    for node in ast.walk(tree):
        kind = type(node).__name__
        if kind not in _ALLOWED_NODES:
            raise RateExprError(
                f"rate expression uses disallowed syntax {kind!r}")
        if isinstance(node, ast.Name) and node.id not in _ALLOWED_NAMES:
            raise RateExprError(
                f"rate expression uses unknown name {node.id!r}")
        if isinstance(node, ast.Attribute):
            # only one attribute level: math.X / np.X, no dunders
            if not (isinstance(node.value, ast.Name)
                    and node.value.id in ("math", "np")
                    and not node.attr.startswith("_")):
                raise RateExprError(
                    "rate expressions may only access math.* / np.*")
        if isinstance(node, ast.Constant) and \
                not isinstance(node.value, (int, float, complex)):
            raise RateExprError("rate expression literals must be numeric")
    code = compile(tree, "<rate-expr>", "eval")
    if len(_EXPR_CACHE) < 4096:
        _EXPR_CACHE[func_string] = code
    return code


def _eval_rate(func_string: str, t: float) -> float:
    """Evaluate an arrival-rate expression f(t) (e.g. 'math.sin(t)+1')."""
    code = compile_rate_expr(func_string)
    return float(eval(code, dict(_SAFE_EVAL_GLOBALS), {"t": t}))



def _dct(v):
    """None/garbage-tolerant dict access: user strategy JSON may carry
    null or mistyped sections (the public analysers never crash —
    fuzz-tested)."""
    return v if isinstance(v, dict) else {}

class Strategy:
    """Pure schedule analysis; no transport attached."""

    # ------------------------------------------------------------ real-time
    @staticmethod
    def check_real_time_dispatch(strategy: str) -> bool:
        s = json.loads(strategy)
        return bool(_dct(s.get("real_time_dispatch")).get("use_strategy", False))

    @staticmethod
    def real_time_strategy_analysis(strategy: str) -> Tuple[List[int], float]:
        s = json.loads(strategy)
        rt = _dct(s.get("real_time_dispatch"))
        batch_sizes = rt.get("dispatch_batch_sizes", [])
        drop_p = _dct(rt.get("drop_simulation")).get("drop_probability", 0)
        return batch_sizes, drop_p

    # ----------------------------------------------------------------- flow
    @classmethod
    def flow_strategy_analysis(cls, strategy: str, flow_id: str,
                               rng: Optional[_random.Random] = None,
                               now: Optional[datetime] = None
                               ) -> Tuple[List[float], List[int], List[List[int]]]:
        s = json.loads(strategy)
        flow = _dct(s.get("flow_dispatch"))
        if not flow.get("use_strategy", False):
            return [], [], []
        total = flow.get("total_dispatch_amount", 0)
        if total <= 0:
            return [], [], []
        use_timing = _dct(flow.get("specific_timing")).get("use", False)
        use_interval = _dct(flow.get("specific_interval")).get("use", False)
        if use_timing == use_interval:          # both or neither -> invalid
            return [], [], []
        inst = cls()
        if use_timing:
            return inst._specific_timing(_dct(flow.get("specific_timing")),
                                         flow_id, rng, now)
        return inst._specific_interval(total, _dct(flow.get("specific_interval")),
                                       flow_id, rng, now)

    # -- explicit send times/amounts ------------------------------------
    def _specific_timing(self, spec: Dict[str, Any], flow_id: str,
                         rng: Optional[_random.Random],
                         now: Optional[datetime]):
        time_type = spec.get("time_type", "relative")
        if time_type == "relative":
            timings = spec.get("timings", [])
        else:
            # absolute timings are per-round lists; the round index is the
            # suffix of flow_id = f"{task}_{operator}_{round}"
            try:
                current_round = int(flow_id.rsplit("_", 1)[1])
                timings = spec.get("timings", [])[current_round]
            except Exception:
                return [], [], []

        amounts = [int(a) for a in spec.get("amounts", [])]
        if len(timings) != len(amounts) or len(timings) == 0:
            return [], [], []

        drop_spec = _dct(spec.get("drop_simulation"))
        if drop_spec:
            if len(drop_spec) != 1:
                return [], [], []
            drops = self.generate_drop_list(amounts, drop_spec, rng)
        else:
            drops = [[] for _ in amounts]

        if time_type == "absolute":
            fmt = "%Y-%m-%d %H:%M:%S"
            current = now or _zone_now(spec)
            frac = current.microsecond / 1e6
            cur = datetime.strptime(current.strftime(fmt), fmt)
            abs_secs = [(datetime.strptime(t, fmt) - cur).total_seconds()
                        for t in timings]
            order = sorted(range(len(abs_secs)), key=lambda i: abs_secs[i])
            abs_secs = [abs_secs[i] for i in order]
            amounts = [amounts[i] for i in order]
            drops = [drops[i] for i in order]
            first_future = next((i for i, t in enumerate(abs_secs) if t >= 0), -1)
            if first_future < 0:
                return [], [], []
            abs_secs = abs_secs[first_future:]
            amounts = amounts[first_future:]
            drops = drops[first_future:]
            timings = [abs_secs[0] - round(frac, 2)] + [
                abs_secs[i] - abs_secs[i - 1] for i in range(1, len(abs_secs))]
        return list(timings), amounts, drops

    # -- arrival-rate functions over intervals ---------------------------
    def _specific_interval(self, total: int, spec: Dict[str, Any],
                           flow_id: str, rng: Optional[_random.Random],
                           now: Optional[datetime]):
        time_type = spec.get("time_type", "relative")
        if time_type == "relative":
            intervals = spec.get("intervals", [])
        else:
            try:
                current_round = int(flow_id.rsplit("_", 1)[1])
                intervals = spec.get("intervals", [])[current_round]
            except Exception:
                return [], [], []

        rules = _dct(spec.get("dispatch_rules"))
        domains = rules.get("domains", [])
        functions = rules.get("functions", [])
        drop_spec = _dct(spec.get("drop_simulation"))
        if len(intervals) != len(domains) or len(domains) != len(functions):
            return [], [], []
        if len(intervals) == 0:
            return [], [], []
        if drop_spec and len(drop_spec) != 1:
            return [], [], []

        if time_type == "absolute":
            fmt = "%Y-%m-%d %H:%M:%S"
            abs_intervals = intervals
            rel: List[List[int]] = []
            for i, (s0, s1) in enumerate(abs_intervals):
                t0 = datetime.strptime(s0, fmt)
                t1 = datetime.strptime(s1, fmt)
                if i == 0:
                    lo = 0
                else:
                    prev_end = datetime.strptime(abs_intervals[i - 1][1], fmt)
                    lo = int((t0 - prev_end).total_seconds()) + rel[i - 1][1]
                rel.append([lo, int((t1 - t0).total_seconds()) + lo])
            timing, amounts, drops = self._interval_schedule(
                total, rel, domains, functions, dict(drop_spec), rng)
            if not timing:
                return [], [], []
            current = now or _zone_now(spec)
            frac = current.microsecond / 1e6
            cur = datetime.strptime(current.strftime(fmt), fmt)
            start = datetime.strptime(abs_intervals[0][0], fmt)
            timing[0] = int((start - cur).total_seconds()) - round(frac, 2)
            # drop slots already in the past
            cum = [timing[0]]
            for d in timing[1:]:
                cum.append(cum[-1] + d)
            first_future = next((i for i, t in enumerate(cum) if t >= 0), -1)
            if first_future < 0:
                return [], [], []
            if first_future > 0:
                timing = timing[first_future:]
                amounts = amounts[first_future:]
                drops = drops[first_future:]
                timing[0] = cum[first_future]
            return timing, amounts, drops

        return self._interval_schedule(total, intervals, domains, functions,
                                       dict(drop_spec), rng)

    def _interval_schedule(self, total: int, intervals, domains, functions,
                           drop_spec: Dict[str, Any],
                           rng: Optional[_random.Random]):
        slot_times: List[List[int]] = []       # send instant per 1-s slot
        slot_areas: List[List[float]] = []     # integrated area per slot
        for interval, domain, func in zip(intervals, domains, functions):
            ilen = interval[1] - interval[0]
            dlen = domain[1] - domain[0]
            ticks = list(range(interval[0], interval[1] + 1))
            dom_ticks = [domain[0] + dlen / ilen * (t - ticks[0]) for t in ticks]
            areas = []
            for i in range(len(dom_ticks) - 1):
                ts = np.linspace(dom_ticks[i], dom_ticks[i + 1],
                                 num=AREA_CALCULATION_NUM + 1)
                ys = [_eval_rate(func, float(t)) for t in ts]
                area = 0.0
                for j in range(1, len(ys)):
                    piece = 0.5 * (ys[j] + ys[j - 1]) * (1.0 / AREA_CALCULATION_NUM)
                    if piece > 0:
                        area += piece
                areas.append(area)
            slot_times.append(ticks[:-1])
            slot_areas.append(areas)

        per_interval_area = [sum(a) for a in slot_areas]
        total_area = sum(per_interval_area)
        if total_area <= 0:
            return [], [], []

        per_interval_amount = [round(a / total_area * total)
                               for a in per_interval_area]
        per_interval_amount[-1] = total - sum(per_interval_amount[:-1])

        sends: List[List[int]] = []
        for idx, amount in enumerate(per_interval_amount):
            if per_interval_area[idx] > 0:
                raw = [a / per_interval_area[idx] * amount
                       for a in slot_areas[idx]]
            else:
                raw = [0.0 for _ in slot_areas[idx]]
            out, carry = [], 0.0
            for r in raw:
                tmp = carry + r
                if round(tmp) > 0:
                    out.append(int(round(tmp)))
                    carry = tmp - round(tmp)
                else:
                    out.append(0)
                    carry = tmp
            sends.append(out)

        # expand interval-level drop settings to slot level
        if "drop_probability" in drop_spec:
            probs = drop_spec.get("drop_probability", [])
            drop_spec["drop_probability"] = [
                probs[i] for i, s in enumerate(sends) for _ in s]
        elif "drop_amounts" in drop_spec:
            amounts_cfg = drop_spec.get("drop_amounts", [])
            expanded: List[int] = []
            rnd = rng or _random
            for i, slots in enumerate(sends):
                slot_sum = sum(slots)
                want = amounts_cfg[i]
                if want == 0:
                    expanded.extend([0] * len(slots))
                elif want == slot_sum:
                    expanded.extend(slots)
                elif 0 < want < slot_sum:
                    chosen = sorted(rnd.sample(range(slot_sum), want))
                    pos = -1
                    for s in slots:
                        hit = 0
                        for _ in range(s):
                            pos += 1
                            if pos in chosen:
                                hit += 1
                        expanded.append(hit)
            drop_spec["drop_amounts"] = expanded

        times_flat = [t for ts in slot_times for t in ts]
        amounts_flat = [a for s in sends for a in s]
        timing = [times_flat[0]] + [times_flat[i] - times_flat[i - 1]
                                    for i in range(1, len(times_flat))]
        if drop_spec:
            drops = self.generate_drop_list(amounts_flat, drop_spec, rng)
        else:
            drops = [[] for _ in amounts_flat]
        return timing, amounts_flat, drops

    # -- drop sampling ----------------------------------------------------
    @staticmethod
    def generate_drop_list(amounts: Sequence[int],
                           drop_spec: Dict[str, Any],
                           rng: Optional[_random.Random] = None
                           ) -> List[List[int]]:
        """Per-slot sorted indices of messages to drop
        (reference _generate_drop_simulation_list, strategy.py:275-311)."""
        rnd = rng or _random
        out: List[List[int]] = []
        if "drop_probability" in drop_spec:
            for p, amount in zip(drop_spec["drop_probability"], amounts):
                if p == 0:
                    out.append([])
                elif p == 1:
                    out.append(list(range(int(amount))))
                elif 0 < p < 1:
                    out.append([i for i in range(int(amount))
                                if rnd.random() < p])
                else:
                    out.append([])
            return out
        if "drop_amounts" in drop_spec:
            for want, amount in zip(drop_spec["drop_amounts"], amounts):
                if want == 0:
                    out.append([])
                elif 0 < want < int(amount):
                    out.append(sorted(rnd.sample(range(int(amount)), int(want))))
                else:
                    out.append(list(range(int(amount))))
            return out
        return []
