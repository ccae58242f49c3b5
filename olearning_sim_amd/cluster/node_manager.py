"""Node/cluster manager — the execution-fabric CRUD.

The reference manages KubeRay clusters through the Kubernetes
CustomObjects API (ols_core/rayclusterMgr/kuberay_cluster_manager.py:
10-225, kuberay_cluster_api.py:28-304, builder/utils): create a
cluster, scale worker groups, delete, query status.  On one 8xMI355X
node the execution fabric is a group of one-process-per-GPU engine
workers launched under torch.distributed; this manager offers the same
CRUD surface over those worker groups:

- create_cluster(spec): records the group and (optionally) launches the
  worker processes via `python -m torch.distributed.run --standalone`;
- update_replicas: scale the group (relaunching);
- delete_cluster: terminate the processes;
- get_cluster / list_clusters / wait_until_running: status queries
  (kuberay_cluster_api.py wait_until_ray_cluster_running analogue).
"""

from __future__ import annotations

import os
import signal
import subprocess
import sys
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class WorkerGroupSpec:
    name: str = "workers"
    replicas: int = 1                 # processes (= GPUs when available)
    entry_module: str = ""            # python module to run per worker
    args: List[str] = field(default_factory=list)
    env: Dict[str, str] = field(default_factory=dict)


@dataclass
class ClusterRecord:
    name: str
    spec: WorkerGroupSpec
    proc: Optional[subprocess.Popen] = None
    created_at: float = 0.0

    def status(self) -> str:
        if self.proc is None:
            return "registered"
        rc = self.proc.poll()
        if rc is None:
            return "running"
        return "succeeded" if rc == 0 else "failed"


class NodeClusterManager:
    def __init__(self):
        self._clusters: Dict[str, ClusterRecord] = {}

    def create_cluster(self, spec: WorkerGroupSpec,
                       launch: bool = True) -> bool:
        if spec.name in self._clusters:
            return False
        rec = ClusterRecord(name=spec.name, spec=spec,
                            created_at=time.time())
        if launch and spec.entry_module:
            rec.proc = self._launch(spec)
        self._clusters[spec.name] = rec
        return True

    @staticmethod
    def _launch(spec: WorkerGroupSpec) -> subprocess.Popen:
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--standalone", "--local-addr", "127.0.0.1",
               f"--nproc-per-node={spec.replicas}",
               "-m", spec.entry_module] + list(spec.args)
        env = dict(os.environ, **spec.env)
        env.setdefault("MASTER_ADDR", "127.0.0.1")
        return subprocess.Popen(cmd, env=env,
                                start_new_session=True)

    def update_replicas(self, name: str, replicas: int) -> bool:
        rec = self._clusters.get(name)
        if rec is None or replicas < 0:
            return False
        rec.spec.replicas = replicas
        if rec.proc is not None:
            self._terminate(rec)
            rec.proc = self._launch(rec.spec) if replicas > 0 else None
        return True

    def delete_cluster(self, name: str) -> bool:
        rec = self._clusters.pop(name, None)
        if rec is None:
            return False
        self._terminate(rec)
        return True

    @staticmethod
    def _terminate(rec: ClusterRecord) -> None:
        if rec.proc is not None and rec.proc.poll() is None:
            # kill the exact process group we started, never by pattern
            try:
                os.killpg(rec.proc.pid, signal.SIGTERM)
                rec.proc.wait(timeout=10)
            except Exception:
                try:
                    os.killpg(rec.proc.pid, signal.SIGKILL)
                except Exception:
                    pass

    def get_cluster(self, name: str) -> Optional[Dict[str, Any]]:
        rec = self._clusters.get(name)
        if rec is None:
            return None
        return {"name": rec.name, "replicas": rec.spec.replicas,
                "status": rec.status(), "created_at": rec.created_at,
                "entry_module": rec.spec.entry_module}

    def list_clusters(self) -> List[str]:
        return list(self._clusters)

    def wait_until_running(self, name: str, timeout: float = 30.0) -> bool:
        t0 = time.time()
        while time.time() - t0 < timeout:
            rec = self._clusters.get(name)
            if rec is None:
                return False
            if rec.status() in ("running", "succeeded"):
                return True
            time.sleep(0.2)
        return False

    def shutdown(self) -> None:
        for name in list(self._clusters):
            self.delete_cluster(name)
