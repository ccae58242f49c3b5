"""gRPC wire endpoint for the six control-plane services.

Serves the reference's exact RPC surface over real protobuf wire
format: taskService.proto's TaskMgr (reference taskService.proto:
205-211), resourceService.proto's ResourceMgr, rayclusterService.proto's
RayClusterMgr, deviceflow.proto's TaskOperatorOrientedDeviceFlow,
performanceService.proto's PerformanceMgr, and phoneMgr.proto's
TaskManager (answered by the simulated phone farm).  The message
classes are compiled at runtime from the .proto contract files by
miniproto.py (no protoc in the image), so a client generated from the
reference's protos with stock protoc interoperates byte-for-byte.

Handlers delegate to the same session facades the JSON/HTTP routes use
(api/server.py); the JSON route map remains the primary documented
transport (docs/PARITY.md) — this endpoint is the wire-compat layer.
"""

from __future__ import annotations

import json
import os
from typing import Dict, Optional

import grpc
from google.protobuf import descriptor_pool, empty_pb2

from .miniproto import load_proto

_PROTO_ORDER = ["taskService.proto", "phoneMgr.proto",
                "resourceService.proto", "deviceflow.proto",
                "rayclusterService.proto", "performanceService.proto"]

_REGISTRY = None


class ProtoRegistry:
    def __init__(self):
        self.pool = descriptor_pool.DescriptorPool()
        base = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                            "protos")
        self.files = {}
        for fname in _PROTO_ORDER:
            self.files[fname] = load_proto(os.path.join(base, fname),
                                           pool=self.pool)

    def msg(self, fname: str, name: str) -> type:
        return self.files[fname].message_class(name)


def registry() -> ProtoRegistry:
    global _REGISTRY
    if _REGISTRY is None:
        _REGISTRY = ProtoRegistry()
    return _REGISTRY


_TRANSFER = {0: "FILE", 1: "HTTP", 2: "S3", 3: "MINIO"}
# TaskStatusEnum numbers (taskService.proto) by our status name
_STATUS_NUM = {"SUCCEEDED": 0, "PENDING": 1, "RUNNING": 2, "STOPPED": 3,
               "FAILED": 4, "MISSING": 5, "UNDONE": 6, "QUEUED": 7}


def _strategy_json(sc) -> Dict:
    return {"strategy": sc.strategyCondition,
            "wait_interval": sc.waitInterval,
            "total_timeout": sc.totalTimeout}


def _opsim_json(si) -> Dict:
    return {"operator_transfer_type": _TRANSFER.get(si.operatorTransferType,
                                                    "FILE"),
            "operator_code_path": si.operatorCodePath,
            "operator_entry_file": si.operatorEntryFile,
            "operator_params": si.operatorParams}


def taskconfig_msg_to_json(msg) -> Dict:
    """TaskConfig wire message -> the canonical task JSON
    (the field mapping of the reference's taskconfig2json,
    ols_core/taskMgr/utils/utils.py:1029-1197)."""
    jd: Dict = {
        "user_id": msg.userID,
        "task_id": msg.taskID.taskID,
        "target": {
            "priority": msg.target.priority,
            "data": [{
                "name": d.dataName,
                "data_path": d.dataPath,
                "data_split_type": d.dataSplitType,
                "data_transfer_type": _TRANSFER.get(d.dataTransferType,
                                                    "FILE"),
                "task_type": d.taskType,
                "total_simulation": {
                    "devices": list(d.totalSimulation.deviceTotalSimulation),
                    "nums": list(d.totalSimulation.numTotalSimulation),
                    "dynamic_nums": list(
                        d.totalSimulation.dynamicNumTotalSimulation)},
                "allocation": {
                    "optimization": d.allocation.optimization,
                    "logical_simulation": list(
                        d.allocation.allocationLogicalSimulation),
                    "device_simulation": list(
                        d.allocation.allocationDeviceSimulation),
                    "running_response": {
                        "devices": list(
                            d.allocation.runningResponse.deviceRunningResponse),
                        "nums": list(
                            d.allocation.runningResponse.numRunningResponse)}},
            } for d in msg.target.targetData]},
        "operatorflow": {
            "flow_setting": {
                "round": msg.operatorFlow.flowSetting.round,
                "start": {
                    "logical_simulation": _strategy_json(
                        msg.operatorFlow.flowSetting.startCondition
                        .logicalSimulationStrategy),
                    "device_simulation": _strategy_json(
                        msg.operatorFlow.flowSetting.startCondition
                        .deviceSimulationStrategy)},
                "stop": {
                    "logical_simulation": _strategy_json(
                        msg.operatorFlow.flowSetting.stopCondition
                        .logicalSimulationStrategy),
                    "device_simulation": _strategy_json(
                        msg.operatorFlow.flowSetting.stopCondition
                        .deviceSimulationStrategy)},
            },
            "operators": [{
                "name": op.name,
                "operation_behavior_controller": {
                    "use_gradient_house":
                        op.operationBehaviorController.useController,
                    "strategy_gradient_house":
                        op.operationBehaviorController
                        .strategyBehaviorController,
                    "outbound_service":
                        op.operationBehaviorController.outboundService},
                "input": list(op.input),
                "use_data": op.useData,
                "model": {
                    "use_model": op.model.useModel,
                    "model_for_train": op.model.modelForTrain,
                    "model_transfer_type": _TRANSFER.get(
                        op.model.modelTransferType, "FILE"),
                    "model_path": op.model.modelPath,
                    "model_update_style": op.model.modelUpdateStyle},
                "logical_simulation": _opsim_json(
                    op.logicalSimulationOperatorInfo),
                "device_simulation": _opsim_json(
                    op.deviceSimulationOperatorInfo),
            } for op in msg.operatorFlow.operator]},
        "logical_simulation": {
            "computation_unit": {
                "devices": list(
                    msg.logicalSimulation.computationUnit.devicesUnit),
                "setting": [{"num_cpus": s.numCpus} for s in
                            msg.logicalSimulation.computationUnit
                            .unitSetting]},
            "resource_request": [{
                "name": r.dataNameResourceRequest,
                "devices": list(r.deviceResourceRequest),
                "num_request": list(r.numResourceRequest)}
                for r in
                msg.logicalSimulation.resourceRequestLogicalSimulation]},
        "device_simulation": {
            "resource_request": [{
                "name": r.dataNameResourceRequest,
                "devices": list(r.deviceResourceRequest),
                "num_request": list(r.numResourceRequest)}
                for r in
                msg.deviceSimulation.resourceRequestDeviceSimulation]},
    }
    return jd


def taskconfig_json_to_msg(raw: Dict):
    """Canonical task JSON -> TaskConfig wire message (client-side
    mirror of the reference's json2taskconfig, utils.py:831-1027).
    Used by wire clients (examples/grpc_client.py, tests)."""
    reg = registry()
    TC = reg.msg("taskService.proto", "TaskConfig")
    m = TC()
    m.userID = raw.get("user_id", "")
    m.taskID.taskID = raw.get("task_id", "")
    tr = {"FILE": 0, "HTTP": 1, "S3": 2, "MINIO": 3}
    tgt = raw.get("target", {})
    m.target.priority = tgt.get("priority", 0)
    for d in tgt.get("data", []):
        td = m.target.targetData.add()
        td.dataName = d.get("name", "")
        td.dataPath = d.get("data_path", "")
        td.dataSplitType = bool(d.get("data_split_type", False))
        td.dataTransferType = tr.get(d.get("data_transfer_type", "FILE"), 0)
        td.taskType = d.get("task_type", "")
        ts_ = d.get("total_simulation", {})
        td.totalSimulation.deviceTotalSimulation.extend(
            ts_.get("devices", []))
        td.totalSimulation.numTotalSimulation.extend(ts_.get("nums", []))
        td.totalSimulation.dynamicNumTotalSimulation.extend(
            ts_.get("dynamic_nums", []))
        al = d.get("allocation", {})
        td.allocation.optimization = bool(al.get("optimization", False))
        td.allocation.allocationLogicalSimulation.extend(
            al.get("logical_simulation", []))
        td.allocation.allocationDeviceSimulation.extend(
            al.get("device_simulation", []))
        rr = al.get("running_response", {})
        td.allocation.runningResponse.deviceRunningResponse.extend(
            rr.get("devices", []))
        td.allocation.runningResponse.numRunningResponse.extend(
            rr.get("nums", []))
    of = raw.get("operatorflow", {})
    fs = of.get("flow_setting", {})
    m.operatorFlow.flowSetting.round = fs.get("round", 1)

    def _cond(dst, src):
        dst.strategyCondition = src.get("strategy", "")
        dst.waitInterval = int(src.get("wait_interval", 0) or 0)
        dst.totalTimeout = int(src.get("total_timeout", 0) or 0)

    start = fs.get("start", {})
    stop = fs.get("stop", {})
    _cond(m.operatorFlow.flowSetting.startCondition
          .logicalSimulationStrategy, start.get("logical_simulation", {}))
    _cond(m.operatorFlow.flowSetting.startCondition
          .deviceSimulationStrategy, start.get("device_simulation", {}))
    _cond(m.operatorFlow.flowSetting.stopCondition
          .logicalSimulationStrategy, stop.get("logical_simulation", {}))
    _cond(m.operatorFlow.flowSetting.stopCondition
          .deviceSimulationStrategy, stop.get("device_simulation", {}))
    for op_raw in of.get("operators", []):
        op = m.operatorFlow.operator.add()
        op.name = op_raw.get("name", "")
        bc = op_raw.get("operation_behavior_controller", {})
        op.operationBehaviorController.useController = bool(
            bc.get("use_gradient_house", False))
        op.operationBehaviorController.strategyBehaviorController = \
            bc.get("strategy_gradient_house", "")
        op.operationBehaviorController.outboundService = \
            bc.get("outbound_service", "")
        op.input.extend(op_raw.get("input", []) or [])
        op.useData = bool(op_raw.get("use_data", False))
        mj = op_raw.get("model", {})
        op.model.useModel = bool(mj.get("use_model", False))
        op.model.modelForTrain = bool(mj.get("model_for_train", False))
        op.model.modelTransferType = tr.get(
            mj.get("model_transfer_type", "S3"), 2)
        op.model.modelPath = mj.get("model_path", "")
        op.model.modelUpdateStyle = mj.get("model_update_style", "")
        for attr, key in ((op.logicalSimulationOperatorInfo,
                           "logical_simulation"),
                          (op.deviceSimulationOperatorInfo,
                           "device_simulation")):
            si = op_raw.get(key, {})
            attr.operatorTransferType = tr.get(
                si.get("operator_transfer_type", "S3"), 2)
            attr.operatorCodePath = si.get("operator_code_path", "")
            attr.operatorEntryFile = si.get("operator_entry_file", "")
            attr.operatorParams = si.get("operator_params", "")
    lsim = raw.get("logical_simulation", {})
    cu = lsim.get("computation_unit", {})
    m.logicalSimulation.computationUnit.devicesUnit.extend(
        cu.get("devices", []))
    for s in cu.get("setting", []):
        m.logicalSimulation.computationUnit.unitSetting.add().numCpus = \
            s.get("num_cpus", 0)
    for r in lsim.get("resource_request", []):
        rr2 = m.logicalSimulation.resourceRequestLogicalSimulation.add()
        rr2.dataNameResourceRequest = r.get("name", "")
        rr2.deviceResourceRequest.extend(r.get("devices", []))
        rr2.numResourceRequest.extend(r.get("num_request", []))
    for r in raw.get("device_simulation", {}).get("resource_request", []):
        rr3 = m.deviceSimulation.resourceRequestDeviceSimulation.add()
        rr3.dataNameResourceRequest = r.get("name", "")
        rr3.deviceResourceRequest.extend(r.get("devices", []))
        rr3.numResourceRequest.extend(r.get("num_request", []))
    return m


def _unary(fn, req_cls):
    deser = (empty_pb2.Empty.FromString if req_cls is None
             else req_cls.FromString)
    return grpc.unary_unary_rpc_method_handler(
        fn, request_deserializer=deser,
        response_serializer=lambda m: m.SerializeToString())


def build_grpc_server(session, port: int = 0,
                      host: str = "127.0.0.1") -> grpc.Server:
    """Build (not start) a grpc.Server with every service the session
    hosts.  Returns the server; the chosen port is server._ols_port."""
    from concurrent import futures
    reg = registry()
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))

    def add(service_full_name: str, methods: Dict[str, tuple]) -> None:
        handlers = {name: _unary(fn, req) for name, (fn, req) in
                    methods.items()}
        server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler(service_full_name,
                                                  handlers),))

    ts = lambda n: reg.msg("taskService.proto", n)          # noqa: E731
    df_ = lambda n: reg.msg("deviceflow.proto", n)          # noqa: E731
    rs = lambda n: reg.msg("resourceService.proto", n)      # noqa: E731
    rc = lambda n: reg.msg("rayclusterService.proto", n)    # noqa: E731
    pm_ = lambda n: reg.msg("phoneMgr.proto", n)            # noqa: E731

    # -- TaskMgr ---------------------------------------------------------
    if session.task_mgr is not None:
        tm = session.task_mgr
        OpStatus = ts("OperationStatus")
        TaskStatusM = ts("TaskStatus")
        TaskQueueM = ts("TaskQueue")

        def submit_task(req, ctx):
            ok, msg = tm.submit_task(json.dumps(taskconfig_msg_to_json(req)))
            if not ok:
                ctx.set_details(msg or "rejected")
            return OpStatus(is_success=ok)

        def stop_task(req, ctx):
            ok, _ = tm.stop_task(req.taskID)
            return OpStatus(is_success=ok)

        def get_task_status(req, ctx):
            st = tm.get_task_status(req.taskID)
            return TaskStatusM(taskStatus=_STATUS_NUM.get(st.name, 5))

        def get_task_queue(req, ctx):
            q = TaskQueueM()
            for tid in tm.get_task_queue():
                q.tasks.add().taskID = tid
            return q

        def change_scheduler(req, ctx):
            return OpStatus(is_success=tm.change_scheduler(req.scheduler))

        add("TaskMgr", {
            "submitTask": (submit_task, ts("TaskConfig")),
            "stopTask": (stop_task, ts("TaskID")),
            "getTaskStatus": (get_task_status, ts("TaskID")),
            "getTaskQueue": (get_task_queue, None),
            "changeScheduler": (change_scheduler, ts("Scheduler")),
        })

    # -- ResourceMgr -----------------------------------------------------
    if session.resource_mgr is not None:
        rm = session.resource_mgr
        SCRes = rs("ServerClusterResource")
        SCStatus = rs("ServerClusterStatus")
        SCDetail = rs("ServerClusterDetail")
        ResourceM = rs("Resource")
        ReqStatus = rs("RequestStatus")
        RelStatus = rs("ReleaseStatus")
        VMRes = rs("VMClusterResource")
        PhoneRes = rs("PhoneClusterResource")
        ResMsg = rs("ResourceMessage")

        def _avail():
            return rm.get_resource("").get("logical_simulation", {})

        def _phone_avail():
            out = {}
            for user, pool in rm.phone_pool.items():
                used = rm._phone_used(user)
                out[user] = {tier: max(0, total - used.get(tier, 0))
                             for tier, total in pool.items()}
            return out

        def _phones_total():
            return sum(n for tiers in _phone_avail().values()
                       for n in tiers.values())

        def get_avail(req, ctx):
            a = _avail()
            return SCRes(cores=float(a.get("cpu", 0.0)),
                         mem=float(a.get("mem", 0.0)))

        def get_detail(req, ctx):
            return SCDetail(detail=json.dumps(rm.get_resource("")))

        def get_total(req, ctx):
            t = rm.totals
            return SCRes(cores=float(t.get("cpu", 0.0)),
                         mem=float(t.get("mem", 0.0)))

        def request_cluster(req, ctx):
            ok = rm.request_resource(req.taskId, "", cpu=req.cores,
                                     mem=req.mem)
            return SCStatus(status=0 if ok else 1)

        def release_cluster(req, ctx):
            ok = rm.release_resource(req.taskId)
            return SCStatus(status=0 if ok else 4)

        def _phone_avail_msg():
            out = pm_("AllUsersDeviceAvailableResource")()
            for user, tiers in _phone_avail().items():
                u = out.userDeviceAvailableResource.add()
                u.userID = user
                for tier, n in tiers.items():
                    info = u.deviceResourceInfo.add()
                    info.phoneType = tier
                    info.num = int(n)
            return out

        def get_resource(req, ctx):
            a = _avail()
            r = ResourceM()
            r.clusterRes.cores = float(a.get("cpu", 0.0))
            r.clusterRes.mem = float(a.get("mem", 0.0))
            r.phoneRes.CopyFrom(_phone_avail_msg())
            return r

        def request_resource(req, ctx):
            phones = {i.phoneType: i.num
                      for i in req.phoneReq.deviceResourceInfo} or None
            task_id = req.clusterReq.taskId or req.phoneReq.taskID
            ok = rm.request_resource(task_id, req.phoneReq.userID,
                                     cpu=req.clusterReq.cores,
                                     mem=req.clusterReq.mem, phones=phones)
            out = ReqStatus()
            out.clusterStatus.status = 0 if ok else 1
            out.phoneStatus.isSuccess = ok
            return out

        def release_resource(req, ctx):
            ok = rm.release_resource(req.taskId)
            out = RelStatus()
            out.clusterStatus.status = 0 if ok else 4
            out.phoneStatus.isSuccess = ok
            return out

        def get_vm(req, ctx):
            return VMRes(availablePhones=_phones_total())

        def get_phone(req, ctx):
            return PhoneRes(avaiablePhones=_phones_total())

        def get_total_resource(req, ctx):
            t = rm.totals
            out = ResMsg()
            out.server.cores = float(t.get("cpu", 0.0))
            out.server.mem = float(t.get("mem", 0.0))
            out.VM.availablePhones = _phones_total()
            out.phone.avaiablePhones = _phones_total()
            return out

        add("ResourceMgr", {
            "getClusterAvailableResource": (get_avail, None),
            "getClusterResourceDetail": (get_detail, None),
            "requestClusterResource": (request_cluster,
                                       rs("ServerClusterReq")),
            "releaseClusterResource": (release_cluster,
                                       rs("ServerClusterReq")),
            "getClusterTotalResource": (get_total, None),
            "getVMResource": (get_vm, None),
            "getPhoneResource": (get_phone, None),
            "getTotalResource": (get_total_resource, None),
            "getResource": (get_resource, None),
            "requestResource": (request_resource, rs("ResReq")),
            "releaseResource": (release_resource, rs("ResRelease")),
        })

    # -- deviceflow ------------------------------------------------------
    if session.deviceflow is not None:
        df = session.deviceflow
        OpResp = df_("OperationResponse")
        PulsarC = df_("PulsarClient")
        WebS = df_("Websocket")
        TotResp = df_("TotalComputeResourcesResponse")

        def _routing(rk: str):
            # routing_key = f"{task}_{operator}_{round}"
            task_op, _, rnd = rk.rpartition("_")
            task, _, op = task_op.rpartition("_")
            return task, op, int(rnd or 0)

        def notify_start(req, ctx):
            task, op, rnd = _routing(req.routing_key)
            fid = df.notify_start(task or req.task_id, op, rnd,
                                  req.compute_resource, req.strategy)
            return OpResp(is_success=fid is not None)

        def notify_complete(req, ctx):
            task, op, rnd = _routing(req.routing_key)
            ok = df.notify_complete(task or req.task_id, op, rnd,
                                    req.compute_resource)
            return OpResp(is_success=ok)

        def register_task(req, ctx):
            ok = df.register_task(req.task_id,
                                  list(req.total_compute_resources))
            return OpResp(is_success=ok)

        def unregister_task(req, ctx):
            return OpResp(is_success=df.unregister_task(req.task_id))

        def get_total_res(req, ctx):
            out = TotResp()
            out.total_compute_resources.extend(
                df.registry.resources(req.task_id) or [])
            return out

        def check_finished(req, ctx):
            return OpResp(is_success=df.check_dispatch_finished(req.task_id))

        def get_pulsar(req, ctx):
            info = df.inbound_info()
            return PulsarC(url=info.get("url", "inproc://deviceflow"),
                           topic=info.get("topic", "inbound"))

        def get_websocket(req, ctx):
            info = df.outbound_info()
            return WebS(url=info.get("url", "inproc://deviceflow/outbound"))

        add("deviceflow.TaskOperatorOrientedDeviceFlow", {
            "GetDeviceflowPulsarClient": (get_pulsar, None),
            "GetDeviceflowWebsocket": (get_websocket, None),
            "NotifyStart": (notify_start, df_("NotifyRequest")),
            "NotifyComplete": (notify_complete,
                               df_("NofifyCompleteRequest")),
            "RegisterTask": (register_task, df_("RegisterRequest")),
            "UnRegisterTask": (unregister_task, df_("UnRegisterRequest")),
            "GetTotalComputeResources": (get_total_res,
                                         df_("TotalComputeResourcesRequest")),
            "CheckDeviceflowDispatchFinished":
                (check_finished, df_("CheckDeviceflowDispatchRequest")),
        })

    # -- RayClusterMgr (NodeClusterManager-backed) -----------------------
    if session.cluster_mgr is not None:
        cm = session.cluster_mgr
        ActResp = rc("RayClusterActResponse")
        QueryRes = rc("RayClusterQueryResult")

        def create_cluster(req, ctx):
            from ..cluster import WorkerGroupSpec
            ok = cm.create_cluster(WorkerGroupSpec(name="default"),
                                   launch=False)
            return ActResp(ok=ok)

        def delete_cluster(req, ctx):
            return ActResp(ok=cm.delete_cluster(req.ray_label or "default"))

        def modify_cluster(req, ctx):
            name = req.ray_name or req.ray_label or "default"
            return ActResp(ok=cm.update_replicas(name, req.worker_replicas))

        def query_cluster(req, ctx):
            info = cm.get_cluster(req.ray_label or "default")
            return QueryRes(json_data=json.dumps(info or {}))

        add("RayClusterMgr", {
            "createRayCluster": (create_cluster, None),
            "deleteRayCluster": (delete_cluster, rc("RayClusterParam")),
            "modifyRayCluster": (modify_cluster,
                                 rc("RayClusterModifyParam")),
            "queryRayCluster": (query_cluster, rc("RayClusterParam")),
        })

    # -- PerformanceMgr --------------------------------------------------
    if session.performance_mgr is not None and session.task_mgr is not None:
        OpStatus = ts("OperationStatus")
        tm2 = session.task_mgr

        def commit_job(req, ctx):
            ok, _ = tm2.submit_task(json.dumps(taskconfig_msg_to_json(req)))
            return OpStatus(is_success=ok)

        add("PerformanceMgr", {"commitJob": (commit_job, ts("TaskConfig"))})

    # -- phoneMgr.TaskManager (simulated farm) ---------------------------
    if session.task_mgr is not None:
        tm3 = session.task_mgr
        Act = pm_("ActionStatus")
        DevResult = pm_("DeviceTaskResult")

        def phone_submit(req, ctx):
            # the farm side is driven by the logical submit; acknowledge
            return Act(isSuccess=True)

        def phone_status(req, ctx):
            row = tm3.table.get_row(req.taskID)
            out = DevResult()
            if row is None:
                return out
            res = tm3._device_task_result(req.taskID, row)
            out.isFinished = bool(res.get("is_finished", False))
            out.round = int(row.get("device_round") or 0)
            out.maxRound = out.round
            out.operator = row.get("device_operator") or ""
            for entry in res.get("device_result", []):
                st = out.deviceDataStatus.add()
                st.name = entry.get("name", "")
                tgt = entry.get("simulation_target", {})
                st.deviceType.extend(tgt.get("devices", []))
                st.successNum.extend(int(x) for x in
                                     tgt.get("success_num", []))
                st.failedNum.extend(int(x) for x in
                                    tgt.get("failed_num", []))
            return out

        def phone_stop(req, ctx):
            ok, _ = tm3.stop_task(req.taskID)
            return Act(isSuccess=ok)

        add("TaskManager", {
            "submitTask": (phone_submit, ts("DeviceTaskConfig")),
            "getDeviceTaskStatus": (phone_status, ts("TaskID")),
            "stopDevice": (phone_stop, ts("TaskID")),
        })

    bound = server.add_insecure_port(f"{host}:{port}")
    server._ols_port = bound
    return server
