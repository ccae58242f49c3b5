"""Framework configuration.

The reference spreads configuration over an INI file + ten YAMLs
(ols_core/config/: config.conf service endpoints and taskMgr timer
periods, repo_*.yaml MySQL tables, deviceflow_config.yaml,
manager_config.yaml S3/MinIO credentials, task_type_config.yaml,
selection_config.yaml, redis.yaml, ray_cluster.yaml).  Here one YAML
(or defaults) configures the whole node: timer periods keep the
reference's names and defaults (config.conf:37-42), storage roots
replace the DB/broker endpoints, and a device section replaces the Ray
endpoints.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Any, Dict, Optional


@dataclass
class SimulatorConfig:
    # taskMgr timer periods (reference config.conf [taskMgr])
    scheduler_sleep_time: float = 5.0
    release_sleep_time: float = 10.0
    interrupt_sleep_time: float = 300.0
    interrupt_queue_time: float = 3600.0
    interrupt_running_time: float = 172800.0

    # storage roots (replace MySQL/Pulsar/MinIO endpoints)
    data_dir: str = ""
    file_root: str = ""
    checkpoint_dir: str = ""

    # execution
    device: str = ""                  # "" = auto (cuda:0 when available)
    api_host: str = "127.0.0.1"
    api_port: int = 60061             # reference session port

    # simulated phone farm quota pool {user_id: {tier: count}}
    phone_pool: Dict[str, Dict[str, int]] = field(default_factory=dict)

    # deviceflow
    deviceflow_time_scale: float = 1.0

    def timers(self) -> Dict[str, float]:
        return {
            "scheduler_sleep_time": self.scheduler_sleep_time,
            "release_sleep_time": self.release_sleep_time,
            "interrupt_sleep_time": self.interrupt_sleep_time,
            "interrupt_queue_time": self.interrupt_queue_time,
            "interrupt_running_time": self.interrupt_running_time,
        }

    @classmethod
    def load(cls, path: Optional[str] = None) -> "SimulatorConfig":
        """Load from YAML; missing file or keys fall back to defaults."""
        cfg = cls()
        path = path or os.environ.get("OLSIM_CONFIG", "")
        if path and os.path.exists(path):
            import yaml
            with open(path) as f:
                raw: Dict[str, Any] = yaml.safe_load(f) or {}
            for k, v in raw.items():
                if hasattr(cfg, k):
                    setattr(cfg, k, v)
        if not cfg.data_dir:
            cfg.data_dir = os.path.join(os.path.expanduser("~"),
                                        ".olearning_sim_amd")
        if not cfg.file_root:
            cfg.file_root = os.path.join(cfg.data_dir, "files")
        if not cfg.checkpoint_dir:
            cfg.checkpoint_dir = os.path.join(cfg.data_dir, "checkpoints")
        return cfg
