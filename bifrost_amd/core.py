"""Core status queries (reference python/bifrost/core.py surface)."""

from bifrost_amd.libbifrost import _bf

__all__ = ["status_string", "debug_enabled", "cuda_enabled"]


def status_string(status):
    return _bf.bfGetStatusString(int(status)).decode()


def debug_enabled():
    return bool(_bf.bfGetDebugEnabled())


def cuda_enabled():
    """True: the HIP backend provides device ('cuda'-space) support."""
    return bool(_bf.bfGetCudaEnabled())
