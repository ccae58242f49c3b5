"""RFC6455 WebSocket transport (utils/ws.py) + its two consumers: the
deviceflow outbound producer (reference message_producer.py:59-78) and
the operator-flow selection poll (reference operatorflow.py:158-237)."""

import base64
import json
import threading
import time

import pytest

from olearning_sim_amd.utils import ws


def test_ws_roundtrip_text_frames():
    got = []
    done = threading.Event()

    def handler(conn):
        while True:
            msg = conn.recv_text(timeout=5.0)
            if msg is None:
                break
            got.append(msg)
            conn.send_text(f"echo:{msg}")
        done.set()

    srv = ws.WebSocketServer(handler)
    try:
        c = ws.connect(srv.url)
        c.send_text("hello")
        assert c.recv_text(timeout=5.0) == "echo:hello"
        # frame sizes across the 126/65536 length encodings
        big = "x" * 70000
        c.send_text(big)
        assert c.recv_text(timeout=5.0) == "echo:" + big
        mid = "y" * 1000
        c.send_text(mid)
        assert c.recv_text(timeout=5.0) == "echo:" + mid
        c.close()
        assert done.wait(5.0)
        assert got == ["hello", big, mid]
    finally:
        srv.shutdown()


def test_deviceflow_websocket_outbound_producer():
    """Messages dispatched by the gradient house arrive at an external
    ws consumer in the reference's wire shape (base64 message field)."""
    from olearning_sim_amd.deviceflow.service import DeviceFlowService
    received = []
    got_one = threading.Event()

    def handler(conn):
        while True:
            msg = conn.recv_text(timeout=10.0)
            if msg is None:
                return
            received.append(json.loads(msg))
            got_one.set()

    srv = ws.WebSocketServer(handler)
    df = DeviceFlowService(":memory:", time_scale=100.0)
    try:
        df.register_task("t_ws", ["logical_simulation"])
        fid = df.notify_start(
            "t_ws", "train", 0, "logical_simulation",
            strategy=json.dumps({"real_time_dispatch": {
                "use_strategy": True, "dispatch_batch_sizes": [1]}}),
            outbound_service=srv.url)
        assert fid is not None
        df.publish(fid, "logical_simulation", payload={"grad": [1, 2]})
        df.drain_inbound()
        df.notify_complete("t_ws", "train", 0, "logical_simulation")
        assert got_one.wait(10.0), "no ws delivery"
        m = received[0]
        assert m["routing_key"] == fid
        assert m["compute_resource"] == "logical_simulation"
        decoded = json.loads(base64.b64decode(m["message"]))
        assert decoded == {"grad": [1, 2]}
    finally:
        df.shutdown()
        srv.shutdown()


def test_operatorflow_ws_selection_gate():
    """waiting_for_global_aggregation over a WebSocket selection
    service: the start gate opens when the service reaches the round."""
    from olearning_sim_amd.engine.operatorflow import OperatorFlow
    state = {"round": -1}

    def handler(conn):
        while True:
            msg = conn.recv_text(timeout=10.0)
            if msg is None:
                return
            q = json.loads(msg)
            assert q["query"] == "round_idx"
            conn.send_text(json.dumps({"round_idx": state["round"]}))

    srv = ws.WebSocketServer(handler)
    try:
        flow = OperatorFlow("t_sel",
                            start_strategy="waiting_for_global_aggregation",
                            wait_interval=0.02, total_timeout=10.0,
                            selection_ws_url=srv.url)

        def advance():
            time.sleep(0.15)
            state["round"] = 0

        t = threading.Thread(target=advance)
        t.start()
        t0 = time.time()
        flow.start(0)            # blocks until the service reaches round 0
        assert time.time() - t0 >= 0.1
        t.join()
        # stop gate: service must advance PAST the round
        state["round"] = 1
        flow.stop(0)
    finally:
        srv.shutdown()
