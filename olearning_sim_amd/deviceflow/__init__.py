from .strategy import Strategy
from .sampler import BehaviorSampler
from .service import DeviceFlowService
from .registry import TaskOrientedDeviceFlowRegistry
from .rooms import Message, InboundRoom, ShelfRoom, OutboundRoom
from .dispatcher import Dispatcher
from .validate import ValidateStrategy

__all__ = ["Strategy", "BehaviorSampler", "DeviceFlowService",
           "TaskOrientedDeviceFlowRegistry", "Message", "InboundRoom",
           "ShelfRoom", "OutboundRoom", "Dispatcher", "ValidateStrategy"]
