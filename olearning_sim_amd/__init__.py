"""olearning_sim_amd — MI355X-native federated-learning device simulator.

A from-scratch rebuild of the capabilities of opas-lab/olearning-sim
(reference surveyed in SURVEY.md) for a single 8x MI355X node:
the control plane (task lifecycle, resources, deviceflow behaviour
simulation) runs in-process on SQLite, and the simulated device compute
runs client-batched on the GPUs (HIP/CDNA4 kernels + RCCL over xGMI).
"""

__version__ = "0.1.0"

from . import utils  # noqa: F401

__all__ = ["utils", "__version__", "SimulatorSession", "build_model",
           "EngineJob", "LogicalEngine"]


def __getattr__(name):
    # lazy top-level conveniences (importing torch/fastapi only on use)
    if name == "SimulatorSession":
        from .session import SimulatorSession
        return SimulatorSession
    if name == "build_model":
        from .models import build_model
        return build_model
    if name == "EngineJob":
        from .engine import EngineJob
        return EngineJob
    if name == "LogicalEngine":
        from .engine import LogicalEngine
        return LogicalEngine
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
