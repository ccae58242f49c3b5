"""Per-flow dispatcher: forwards shelved messages per the behaviour
strategy.

Parity with the reference's Dispatcher
(ols_core/deviceflow/non_grpc/dispatcher.py:47-252):

- real-time mode (:84-171): accumulate batches of `dispatch_batch_sizes`
  (cycled), drop each message with `drop_probability`, forward the rest;
  on release, flush whatever remains.
- flow mode (:174-252): the (dispatch_timing, dispatch_amount,
  drop_simulation_list) schedule from Strategy.flow_strategy_analysis —
  wait `timing[i]` seconds, take `amount[i]` messages off the shelf,
  drop the listed in-slot indices, forward the rest; afterwards drain
  leftovers (clean_remain_message).

`time_scale` compresses the waits (tests use 0); the reference sleeps
real seconds.
"""

from __future__ import annotations

import random
import threading
import time
from typing import List, Optional

from .rooms import Message, OutboundRoom, ShelfRoom
from .strategy import Strategy


class Dispatcher:
    def __init__(self, flow_id: str, strategy: str, shelf: ShelfRoom,
                 outbound: OutboundRoom, time_scale: float = 1.0,
                 rng: Optional[random.Random] = None):
        self.flow_id = flow_id
        self.strategy = strategy
        self.shelf = shelf
        self.outbound = outbound
        self.time_scale = time_scale
        self.rng = rng or random.Random()
        self.release_event = threading.Event()   # NotifyComplete fired
        self.stop_event = threading.Event()
        self.finished = threading.Event()
        self.forwarded = 0
        self.dropped = 0
        # per-slot dispatch curve: (t_virtual_s, sent, dropped, cumulative)
        # — the reference's operation_amount_table /
        # accumulated_amount_table demo rows (dispatcher.py:254-395)
        self.history: List[tuple] = []
        self._t_virtual = 0.0
        self._thread: Optional[threading.Thread] = None

    def _record_slot(self, wait_s: float, sent: int, dropped: int) -> None:
        self._t_virtual += wait_s
        self.history.append((self._t_virtual, sent, dropped, self.forwarded))

    # ------------------------------------------------------------------
    def start(self) -> None:
        self._thread = threading.Thread(target=self.dispatch, daemon=True)
        self._thread.start()

    def join(self, timeout: float = 30.0) -> None:
        if self._thread is not None:
            self._thread.join(timeout)

    def dispatch(self) -> None:
        try:
            if Strategy.check_real_time_dispatch(self.strategy):
                self._dispatch_real_time()
            else:
                self._dispatch_flow()
        finally:
            self.finished.set()

    # -- real-time mode --------------------------------------------------
    def _dispatch_real_time(self) -> None:
        batch_sizes, drop_p = Strategy.real_time_strategy_analysis(self.strategy)
        batch_idx = 0
        batch: List[Message] = []
        while not self.stop_event.is_set():
            target = (batch_sizes[batch_idx % len(batch_sizes)]
                      if batch_sizes else 1)
            msgs = self.shelf.take(self.flow_id, target - len(batch),
                                   timeout=0.01)
            batch.extend(msgs)
            if len(batch) >= target and target > 0:
                n0, d0 = self.forwarded, self.dropped
                self._forward_with_drop(batch, drop_p)
                self._record_slot(0.0, self.forwarded - n0, self.dropped - d0)
                batch = []
                batch_idx += 1
            elif self.release_event.is_set() and \
                    self.shelf.depth(self.flow_id) == 0:
                n0, d0 = self.forwarded, self.dropped
                self._forward_with_drop(batch, drop_p)   # flush remainder
                if len(batch):
                    self._record_slot(0.0, self.forwarded - n0,
                                      self.dropped - d0)
                return

    def _forward_with_drop(self, batch: List[Message], drop_p: float) -> None:
        for m in batch:
            if drop_p > 0 and self.rng.random() < drop_p:
                self.dropped += 1
            else:
                self.outbound.send(m)
                self.forwarded += 1

    # -- flow mode --------------------------------------------------------
    def _dispatch_flow(self) -> None:
        # schedule computed once the flow is released (reference: flow
        # strategies dispatch at NotifyComplete)
        self.release_event.wait()
        timing, amounts, drops = Strategy.flow_strategy_analysis(
            self.strategy, self.flow_id, rng=self.rng)
        for wait_s, amount, drop_idx in zip(timing, amounts, drops):
            if self.stop_event.is_set():
                break
            if wait_s > 0 and self.time_scale > 0:
                time.sleep(wait_s * self.time_scale)
            msgs = self.shelf.take(self.flow_id, amount, timeout=0.05)
            dropset = set(drop_idx)
            n0, d0 = self.forwarded, self.dropped
            for i, m in enumerate(msgs):
                if i in dropset:
                    self.dropped += 1
                else:
                    self.outbound.send(m)
                    self.forwarded += 1
            self._record_slot(wait_s, self.forwarded - n0, self.dropped - d0)
        # clean_remain_message (dispatcher.py:244-252)
        for m in self.shelf.take(self.flow_id, 1_000_000):
            self.outbound.send(m)
            self.forwarded += 1
