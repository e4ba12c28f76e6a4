from .client import TokenClient, query_stats
from .local import LocalGPUShare, PodHandle, native_path

__all__ = ["LocalGPUShare", "PodHandle", "TokenClient", "native_path",
           "query_stats"]
