from .dist_context import DeviceMeshParameters, DistributedContext

__all__ = ["DeviceMeshParameters", "DistributedContext"]
