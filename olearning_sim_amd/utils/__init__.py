from .sqlite_repo import SqlTableRepo
from .logging import Logger

__all__ = ["SqlTableRepo", "Logger"]
