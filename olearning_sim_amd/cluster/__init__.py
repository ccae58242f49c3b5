from .node_manager import NodeClusterManager, WorkerGroupSpec

__all__ = ["NodeClusterManager", "WorkerGroupSpec"]
