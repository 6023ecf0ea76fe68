from .late_init import ModuleLateInit

__all__ = ["ModuleLateInit"]
