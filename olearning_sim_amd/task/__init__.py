from .status import TaskStatus
from .schema import (TaskConfig, json2taskconfig, taskconfig2json,
                     FILE_TRANSFER_TYPES)
from .validate import ValidateParameters
from .queue import TaskQueue
from .table import TaskTableRepo

__all__ = [
    "TaskStatus", "TaskConfig", "json2taskconfig", "taskconfig2json",
    "FILE_TRANSFER_TYPES", "ValidateParameters", "TaskQueue", "TaskTableRepo",
]
