from .manager import ResourceManager

__all__ = ["ResourceManager"]
