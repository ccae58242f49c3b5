"""SimulatorSession composition + JSON API surface (reference
simu_session.py + the six proto service surfaces)."""

import json
import time

import pytest
from fastapi.testclient import TestClient

from olearning_sim_amd.session import SimulatorSession
from olearning_sim_amd.api.server import build_app

from test_manager import task_json


@pytest.fixture
def session(tmp_path):
    s = SimulatorSession(svc=0, data_dir=str(tmp_path), device="cpu",
                         auto_start_threads=False)
    yield s
    s.shutdown()


@pytest.fixture
def client(session):
    return TestClient(build_app(session))


def test_health_and_composition(client, session):
    assert client.get("/health").json()["ok"]
    assert session.task_mgr is not None
    assert session.resource_mgr is not None
    assert session.deviceflow is not None
    assert session.performance_mgr is not None


def test_svc_codes(tmp_path):
    s = SimulatorSession(svc=2, data_dir=str(tmp_path / "a"))
    assert s.task_mgr is None and s.resource_mgr is not None
    s.shutdown()
    s = SimulatorSession(svc=4, data_dir=str(tmp_path / "b"))
    assert s.performance_mgr is not None and s.task_mgr is None
    s.shutdown()


def test_submit_and_run_task_via_api(client, session):
    body = {"task": json.loads(task_json(task_id="t_api"))}
    r = client.post("/taskmgr/submitTask", json=body).json()
    assert r["is_success"], r
    assert client.get("/taskmgr/getTaskQueue").json()["tasks"] == ["t_api"]
    assert client.get("/taskmgr/getTaskStatus/t_api").json()[
        "task_status"] == "QUEUED"
    # drive the scheduler manually (threads off in tests)
    assert session.task_mgr.step_schedule() == "t_api"
    t0 = time.time()
    while time.time() - t0 < 30:
        st = client.get("/taskmgr/getTaskStatus/t_api").json()["task_status"]
        if st in ("SUCCEEDED", "FAILED", "STOPPED"):
            break
        time.sleep(0.05)
    assert st == "SUCCEEDED"


def test_invalid_task_rejected_via_api(client):
    bad = json.loads(task_json(task_id="t_bad"))
    bad["target"]["priority"] = 99
    r = client.post("/taskmgr/submitTask", json={"task": bad}).json()
    assert not r["is_success"]


def test_resource_api_cycle(client):
    before = client.get("/resourcemgr/getResource").json()
    assert "logical_simulation" in before
    ok = client.post("/resourcemgr/requestResource", json={
        "task_id": "r1", "cpu": 1.0, "mem": 1.0}).json()["is_success"]
    assert ok
    after = client.get("/resourcemgr/getResource").json()
    assert after["logical_simulation"]["cpu"] == \
        before["logical_simulation"]["cpu"] - 1.0
    assert client.post("/resourcemgr/releaseResource/r1").json()["is_success"]


def test_deviceflow_api_lifecycle(client):
    assert client.post("/deviceflow/RegisterTask", json={
        "task_id": "t", "total_compute_resources": ["logical_simulation"],
    }).json()["is_success"]
    r = client.post("/deviceflow/NotifyStart", json={
        "task_id": "t", "operator_name": "train", "round": 0,
        "compute_resource": "logical_simulation",
        "strategy": json.dumps({"real_time_dispatch": {
            "use_strategy": True, "dispatch_batch_sizes": [1]}})}).json()
    assert r["is_success"] and r["flow_id"] == "t_train_0"
    assert client.post("/deviceflow/NotifyComplete", json={
        "task_id": "t", "operator_name": "train", "round": 0,
        "compute_resource": "logical_simulation"}).json()["is_success"]
    t0 = time.time()
    while time.time() - t0 < 10:
        if client.get("/deviceflow/CheckDeviceflowDispatchFinished/t"
                      ).json()["is_finished"]:
            break
        time.sleep(0.05)
    assert client.post("/deviceflow/UnRegisterTask/t").json()["is_success"]


def test_cluster_api(client):
    assert client.post("/cluster/create", json={
        "name": "grp", "replicas": 2}).json()["is_success"]
    info = client.get("/cluster/get/grp").json()
    assert info["replicas"] == 2 and info["status"] == "registered"
    assert client.post("/cluster/updateReplicas/grp/4").json()["is_success"]
    assert client.get("/cluster/get/grp").json()["replicas"] == 4
    assert client.get("/cluster/list").json()["clusters"] == ["grp"]
    assert client.post("/cluster/delete/grp").json()["is_success"]


def test_performance_api(client, session):
    session.performance_mgr.record_round("t", 0, 1.5, 30, loss=2.0)
    session.performance_mgr.record_round("t", 1, 1.0, 30, loss=1.5)
    s = client.get("/performancemgr/summary/t").json()
    assert s["metrics"]["round_time_s"]["count"] == 2
    assert s["metrics"]["clients_per_s"]["last"] == 30.0
    rows = client.get("/performancemgr/metrics/t",
                      params={"metric": "loss"}).json()["metrics"]
    assert [r["value"] for r in rows] == [2.0, 1.5]


@pytest.mark.timeout(420)
def test_cluster_manager_launches_distributed_worker(tmp_path):
    """NodeClusterManager runs the engine worker group (2 CPU ranks over
    gloo) — the control-plane path to multi-GPU execution."""
    import json as _json
    from olearning_sim_amd.cluster import NodeClusterManager, WorkerGroupSpec
    job = {"task_id": "grp_job", "model_name": "mlp",
           "model_kwargs": {"in_features": 16, "hidden": 8, "num_classes": 4},
           "clients": 8, "rounds": 2, "local_steps": 1, "batch_size": 2,
           "lr": 0.1, "num_classes": 4, "shard_size": 8, "seed": 5,
           "dtype": "float32"}
    job_file = tmp_path / "job.json"
    job_file.write_text(_json.dumps(job))
    result = tmp_path / "result.json"
    cm = NodeClusterManager()
    assert cm.create_cluster(WorkerGroupSpec(
        name="eng", replicas=2,
        entry_module="olearning_sim_amd.engine.worker",
        args=["--job-json", str(job_file), "--result-json", str(result)]))
    assert cm.wait_until_running("eng", timeout=60)
    rec = cm._clusters["eng"]
    rec.proc.wait(timeout=300)
    assert rec.status() == "succeeded"
    out = _json.loads(result.read_text())
    assert out["rounds"] == 2
    assert out["world_size"] == 2
    assert out["success_total"] == 2 * 8   # all shards, both rounds
    cm.delete_cluster("eng")


def test_list_tasks_and_result_api(client, session):
    body = {"task": json.loads(task_json(task_id="t_res"))}
    assert client.post("/taskmgr/submitTask", json=body).json()["is_success"]
    assert session.task_mgr.step_schedule() == "t_res"
    t0 = time.time()
    while time.time() - t0 < 30:
        st = client.get("/taskmgr/getTaskStatus/t_res").json()["task_status"]
        if st in ("SUCCEEDED", "FAILED", "STOPPED"):
            break
        time.sleep(0.05)
    tasks = client.get("/taskmgr/listTasks").json()["tasks"]
    assert any(t["task_id"] == "t_res" for t in tasks)
    res = client.get("/taskmgr/getTaskResult/t_res").json()
    assert res["task_status"] == "SUCCEEDED"
    tgt = res["logical_result"]["logical_result"][0]["simulation_target"]
    assert tgt["success_num"] == [6]
    assert client.get("/taskmgr/getTaskResult/nope").json()["error"]


def test_session_honours_config(tmp_path):
    from olearning_sim_amd.config import SimulatorConfig
    cfg = SimulatorConfig(scheduler_sleep_time=0.7,
                          phone_pool={"u9": {"high": 2}},
                          deviceflow_time_scale=0.0)
    s = SimulatorSession(svc=0, data_dir=str(tmp_path), device="cpu",
                         auto_start_threads=False, config=cfg)
    assert s.task_mgr.timers["scheduler_sleep_time"] == 0.7
    avail = s.resource_mgr.get_resource("u9")["device_simulation"]["u9"]
    assert avail == {"high": 2}
    assert s.deviceflow.time_scale == 0.0
    s.shutdown()


@pytest.mark.timeout(120)
def test_real_http_server_round_trip(tmp_path):
    """serve() over a real socket (uvicorn), not just the TestClient."""
    import socket
    import urllib.request
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    s = SimulatorSession(svc=0, data_dir=str(tmp_path), device="cpu",
                         auto_start_threads=False)
    server = s.serve(host="127.0.0.1", port=port, block=False)
    try:
        t0 = time.time()
        while time.time() - t0 < 30:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/health", timeout=5) as r:
                    body = json.loads(r.read())
                break
            except Exception:
                time.sleep(0.2)
        assert body["ok"]
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/taskmgr/submitTask",
            data=json.dumps({"task": json.loads(
                task_json(task_id="t_http"))}).encode(),
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=10) as r:
            assert json.loads(r.read())["is_success"]
        with urllib.request.urlopen(
                f"http://127.0.0.1:{port}/taskmgr/getTaskQueue",
                timeout=5) as r:
            assert json.loads(r.read())["tasks"] == ["t_http"]
    finally:
        server.should_exit = True
        s.shutdown()


def test_deviceflow_data_plane_over_http(client):
    """Publish -> sorter -> dispatcher -> outbound, all through the JSON
    routes (GetDeviceflowPulsarClient/GetDeviceflowWebsocket analogues
    plus the in-process inbound/outbound endpoints)."""
    import json as _json
    import time as _time
    assert client.get("/deviceflow/GetInboundInfo").json()["kind"] == "inproc"
    assert client.get("/deviceflow/GetOutboundInfo").json()[
        "endpoint"] == "/deviceflow/outbound"

    strategy = _json.dumps({"real_time_dispatch": {
        "use_strategy": True, "dispatch_batch_sizes": [1]}})
    assert client.post("/deviceflow/RegisterTask", json={
        "task_id": "t_dp",
        "total_compute_resources": ["logical_simulation"]}).json()["is_success"]
    fid = client.post("/deviceflow/NotifyStart", json={
        "task_id": "t_dp", "operator_name": "train", "round": 0,
        "compute_resource": "logical_simulation",
        "strategy": strategy}).json()["flow_id"]
    for i in range(3):
        assert client.post("/deviceflow/publish", json={
            "routing_key": fid, "compute_resource": "logical_simulation",
            "payload": {"i": i}}).json()["is_success"]
    t0 = _time.time()
    while _time.time() - t0 < 10:
        if client.get("/deviceflow/GetInboundInfo").json()["queue_depth"] == 0:
            break
        _time.sleep(0.01)
    assert client.post("/deviceflow/NotifyComplete", json={
        "task_id": "t_dp", "operator_name": "train", "round": 0,
        "compute_resource": "logical_simulation"}).json()["is_success"]
    got = []
    t0 = _time.time()
    while _time.time() - t0 < 10 and len(got) < 3:
        got += client.get("/deviceflow/outbound").json()["messages"]
        _time.sleep(0.01)
    assert sorted(m["payload"]["i"] for m in got) == [0, 1, 2]


def test_svc_codes_1_and_3(tmp_path):
    s = SimulatorSession(svc=1, data_dir=str(tmp_path / "c"))
    assert s.task_mgr is not None and s.performance_mgr is None
    s.shutdown()
    s = SimulatorSession(svc=3, data_dir=str(tmp_path / "d"))
    assert s.deviceflow is not None and s.task_mgr is None
    s.shutdown()


def test_phonemgr_device_task_status_route(client, session):
    """The phone-side RPC surface (phoneMgr.proto analogue) reports the
    simulated farm's DeviceTaskResult."""
    body = {"task": json.loads(task_json(task_id="t_phone"))}
    assert client.post("/taskmgr/submitTask", json=body).json()["is_success"]
    # logical-only task: no device side, so the farm reports finished
    r = client.get("/phonemgr/getDeviceTaskStatus/t_phone").json()
    assert r["is_finished"] and r["device_result"] == []
    assert "error" in client.get(
        "/phonemgr/getDeviceTaskStatus/nope").json()
