"""Simulator session: compose the services into one process.

Parity with the reference's SimulatorSession
(ols_core/simu_session.py:25-71): one server process mounts the chosen
services by `svc` code — 0 mounts everything, 1 TaskMgr, 2 ResourceMgr,
3 DeviceFlow, 4 PerformanceMgr (the reference mounts PerformanceMgr for
svc in {0,4}; its RayClusterMgr is commented out, here the
NodeClusterManager is mounted with svc 0).  `serve()` exposes the JSON
API over HTTP (api/server.py) — the reference's gRPC surface — while
in-process use needs no server at all.
"""

from __future__ import annotations

import os
import threading
from typing import Optional

from .cluster import NodeClusterManager
from .deviceflow.service import DeviceFlowService
from .perf import PerformanceManager
from .resource.manager import ResourceManager
from .task.manager import TaskManager
from .task.runner import TaskRunner
from .task.table import TaskTableRepo
from .utils.logging import Logger


class SimulatorSession:
    def __init__(self, svc: int = 0, data_dir: Optional[str] = None,
                 device: str = "", auto_start_threads: bool = True,
                 config=None):
        from .config import SimulatorConfig
        self.config = config or SimulatorConfig.load(None)
        self.svc = svc
        self.data_dir = (data_dir or self.config.data_dir
                         or os.path.join(os.path.expanduser("~"),
                                         ".olearning_sim_amd"))
        os.makedirs(self.data_dir, exist_ok=True)
        self.log = Logger.shared()

        def db(name: str) -> str:
            return os.path.join(self.data_dir, name)

        if not device:
            device = self.config.device
        if not device:
            try:
                import torch
                device = "cuda:0" if torch.cuda.is_available() else "cpu"
            except Exception:
                device = "cpu"

        self.resource_mgr = (ResourceManager(
                                 db("resmgr.sqlite"),
                                 phone_pool=self.config.phone_pool)
                             if svc in (0, 1, 2) else None)
        self.deviceflow = (DeviceFlowService(
                               db("deviceflow.sqlite"),
                               time_scale=self.config.deviceflow_time_scale)
                           if svc in (0, 3) else None)
        self.performance_mgr = (PerformanceManager(db("perf.sqlite"))
                                if svc in (0, 4) else None)
        self.cluster_mgr = NodeClusterManager() if svc in (0, 1) else None
        self.task_mgr = None
        if svc in (0, 1):
            table = TaskTableRepo(db("taskmgr.sqlite"))
            runner = TaskRunner(
                table, device=device,
                checkpoint_dir=os.path.join(self.data_dir, "checkpoints"),
                deviceflow=self.deviceflow, perf=self.performance_mgr,
                cluster=self.cluster_mgr)
            self.task_mgr = TaskManager(
                table=table, resource_mgr=self.resource_mgr, runner=runner,
                deviceflow=self.deviceflow, timers=self.config.timers(),
                auto_start=auto_start_threads)

    def serve(self, host: str = "127.0.0.1", port: int = 60061,
              block: bool = True):
        """Serve the JSON API (reference: grpc server on the session
        port).  Returns the uvicorn server when block=False."""
        import uvicorn
        from .api.server import build_app
        app = build_app(self)
        config = uvicorn.Config(app, host=host, port=port, log_level="warning")
        server = uvicorn.Server(config)
        if block:
            server.run()
            return server
        t = threading.Thread(target=server.run, daemon=True)
        t.start()
        return server

    def serve_grpc(self, host: str = "127.0.0.1", port: int = 0):
        """Serve the six services over real protobuf wire format (the
        reference's gRPC surface; message classes compiled at runtime
        from api/protos/*.proto by api/miniproto.py).  Returns the
        started grpc.Server; the bound port is ``server._ols_port``."""
        from .api.grpc_server import build_grpc_server
        server = build_grpc_server(self, port=port, host=host)
        server.start()
        return server

    def shutdown(self) -> None:
        if self.task_mgr is not None:
            self.task_mgr.shutdown()
        if self.deviceflow is not None:
            self.deviceflow.shutdown()
        if self.cluster_mgr is not None:
            self.cluster_mgr.shutdown()
