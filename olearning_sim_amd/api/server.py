"""JSON API over the simulator session.

The reference exposes six gRPC services (ols_core/proto/*.proto:
TaskMgr taskService.proto:205-211, ResourceMgr
resourceService.proto:75-104, RayClusterMgr rayclusterService.proto:
26-38, PerformanceMgr performanceService.proto:4-6,
TaskOperatorOrientedDeviceFlow deviceflow.proto:63-72, and the
phone-side TaskManager).  gRPC codegen is unavailable offline, so the
same RPC surface is served as JSON-over-HTTP (FastAPI) — rpc name ->
route, request/response messages -> JSON bodies with the task-JSON
schema unchanged.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from fastapi import FastAPI
from pydantic import BaseModel


class SubmitTaskBody(BaseModel):
    task: Dict[str, Any]            # the canonical task JSON


class ResourceRequestBody(BaseModel):
    task_id: str
    user_id: str = ""
    cpu: float = 0.0
    mem: float = 0.0
    gpu: float = 0.0
    hbm_gb: float = 0.0
    phones: Optional[Dict[str, int]] = None


class NotifyBody(BaseModel):
    task_id: str
    operator_name: str
    round: int
    compute_resource: str
    strategy: str = ""
    outbound_service: str = ""


class RegisterBody(BaseModel):
    task_id: str
    total_compute_resources: list


class PublishBody(BaseModel):
    routing_key: str
    compute_resource: str
    payload: Optional[Any] = None


class ClusterBody(BaseModel):
    name: str
    replicas: int = 1
    entry_module: str = ""
    args: list = []


def build_app(session) -> FastAPI:
    app = FastAPI(title="olearning_sim_amd",
                  description="MI355X-native federated-learning device "
                              "simulator control plane")

    # -- TaskMgr service (taskService.proto) -----------------------------
    if session.task_mgr is not None:
        tm = session.task_mgr

        @app.post("/taskmgr/submitTask")
        def submit_task(body: SubmitTaskBody):
            import json
            ok, msg = tm.submit_task(json.dumps(body.task))
            return {"is_success": ok, "message": msg}

        @app.post("/taskmgr/stopTask/{task_id}")
        def stop_task(task_id: str):
            ok, msg = tm.stop_task(task_id)
            return {"is_success": ok, "message": msg}

        @app.get("/taskmgr/getTaskStatus/{task_id}")
        def get_task_status(task_id: str):
            return {"task_status": tm.get_task_status(task_id).value}

        @app.get("/taskmgr/getTaskQueue")
        def get_task_queue():
            return {"tasks": tm.get_task_queue()}

        @app.get("/taskmgr/listTasks")
        def list_tasks():
            rows = tm.table.all_rows()
            return {"tasks": [{
                "task_id": r["task_id"], "user_id": r.get("user_id"),
                "task_status": r.get("task_status"),
                "logical_round": r.get("logical_round"),
                "in_queue_time": r.get("in_queue_time"),
                "finish_task_time": r.get("finish_task_time"),
            } for r in rows]}

        @app.get("/taskmgr/getTaskResult/{task_id}")
        def get_task_result(task_id: str):
            import json as _json
            row = tm.table.get_row(task_id)
            if row is None:
                return {"error": "task not found"}
            return {
                "task_id": task_id,
                "task_status": tm.get_task_status(task_id).value,
                "logical_round": row.get("logical_round"),
                "logical_operator": row.get("logical_operator"),
                "logical_result": _json.loads(row["logical_result"])
                if row.get("logical_result") else None,
                "device_result": _json.loads(row["device_result"])
                if row.get("device_result") else None,
            }

    # -- ResourceMgr service (resourceService.proto) ---------------------
    if session.resource_mgr is not None:
        rm = session.resource_mgr

        @app.get("/resourcemgr/getResource")
        def get_resource(user_id: str = ""):
            return rm.get_resource(user_id)

        @app.post("/resourcemgr/requestResource")
        def request_resource(body: ResourceRequestBody):
            ok = rm.request_resource(
                body.task_id, body.user_id, cpu=body.cpu, mem=body.mem,
                gpu=body.gpu, hbm_gb=body.hbm_gb, phones=body.phones)
            return {"is_success": ok}

        @app.post("/resourcemgr/releaseResource/{task_id}")
        def release_resource(task_id: str):
            return {"is_success": rm.release_resource(task_id)}

    # -- DeviceFlow service (deviceflow.proto) ---------------------------
    if session.deviceflow is not None:
        df = session.deviceflow

        @app.post("/deviceflow/RegisterTask")
        def register_task(body: RegisterBody):
            ok = df.register_task(body.task_id, body.total_compute_resources)
            return {"is_success": ok}

        @app.post("/deviceflow/UnRegisterTask/{task_id}")
        def unregister_task(task_id: str):
            return {"is_success": df.unregister_task(task_id)}

        @app.post("/deviceflow/NotifyStart")
        def notify_start(body: NotifyBody):
            fid = df.notify_start(body.task_id, body.operator_name,
                                  body.round, body.compute_resource,
                                  body.strategy, body.outbound_service)
            return {"is_success": fid is not None, "flow_id": fid}

        @app.post("/deviceflow/NotifyComplete")
        def notify_complete(body: NotifyBody):
            ok = df.notify_complete(body.task_id, body.operator_name,
                                    body.round, body.compute_resource)
            return {"is_success": ok}

        @app.get("/deviceflow/CheckDeviceflowDispatchFinished/{task_id}")
        def check_finished(task_id: str):
            return {"is_finished": df.check_dispatch_finished(task_id)}

        # connection-info RPCs (GetDeviceflowPulsarClient /
        # GetDeviceflowWebsocket analogues) + the in-process data plane
        @app.get("/deviceflow/GetInboundInfo")
        def get_inbound_info():
            return df.inbound_info()

        @app.get("/deviceflow/GetOutboundInfo")
        def get_outbound_info():
            return df.outbound_info()

        @app.get("/deviceflow/dispatchCurve/{task_id}")
        def dispatch_curve(task_id: str):
            return {"flows": df.dispatch_curve(task_id)}

        @app.post("/deviceflow/publish")
        def df_publish(body: PublishBody):
            df.publish(body.routing_key, body.compute_resource, body.payload)
            return {"is_success": True}

        @app.get("/deviceflow/outbound")
        def df_outbound(max_items: int = 100):
            return {"messages": [
                {"routing_key": m.routing_key,
                 "compute_resource": m.compute_resource,
                 "payload": m.payload}
                for m in df.outbound.drain(max_items)]}

    # -- phone-side surface (phoneMgr.proto analogue; the proprietary
    # real-phone farm is simulated by the runner's device jobs) ---------
    if session.task_mgr is not None:
        ptm = session.task_mgr

        @app.get("/phonemgr/getDeviceTaskStatus/{task_id}")
        def get_device_task_status(task_id: str):
            row = ptm.table.get_row(task_id)
            if row is None:
                return {"error": "task not found"}
            return ptm._device_task_result(task_id, row)

    # -- PerformanceMgr service (performanceService.proto) ---------------
    if session.performance_mgr is not None:
        pm = session.performance_mgr

        @app.get("/performancemgr/summary/{task_id}")
        def perf_summary(task_id: str):
            return pm.summary(task_id)

        @app.get("/performancemgr/metrics/{task_id}")
        def perf_metrics(task_id: str, metric: str = ""):
            return {"metrics": pm.metrics(task_id, metric or None)}

    # -- cluster manager (rayclusterService.proto analogue) --------------
    if session.cluster_mgr is not None:
        cm = session.cluster_mgr

        @app.post("/cluster/create")
        def create_cluster(body: ClusterBody):
            from ..cluster import WorkerGroupSpec
            ok = cm.create_cluster(WorkerGroupSpec(
                name=body.name, replicas=body.replicas,
                entry_module=body.entry_module, args=body.args),
                launch=bool(body.entry_module))
            return {"is_success": ok}

        @app.post("/cluster/updateReplicas/{name}/{replicas}")
        def update_replicas(name: str, replicas: int):
            return {"is_success": cm.update_replicas(name, replicas)}

        @app.post("/cluster/delete/{name}")
        def delete_cluster(name: str):
            return {"is_success": cm.delete_cluster(name)}

        @app.get("/cluster/get/{name}")
        def get_cluster(name: str):
            return cm.get_cluster(name) or {"error": "not found"}

        @app.get("/cluster/list")
        def list_clusters():
            return {"clusters": cm.list_clusters()}

    @app.get("/health")
    def health():
        return {"ok": True, "svc": session.svc}

    return app
