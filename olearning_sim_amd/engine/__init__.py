from .job import EngineJob
from .round_loop import LogicalEngine
from .client_manager import FlatParams
from .data import SyntheticFederatedData

__all__ = ["EngineJob", "LogicalEngine", "FlatParams", "SyntheticFederatedData"]
