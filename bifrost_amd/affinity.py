"""CPU core binding helpers (reference python/bifrost/affinity.py surface)."""

import ctypes

from bifrost_amd.libbifrost import _bf, _check

__all__ = ["get_core", "set_core", "set_openmp_cores"]


def get_core():
    core = ctypes.c_int()
    _check(_bf.bfAffinityGetCore(ctypes.byref(core)))
    return core.value


def set_core(core):
    _check(_bf.bfAffinitySetCore(-1 if core is None else int(core)))


def set_openmp_cores(cores):
    arr = (ctypes.c_int * len(cores))(*cores)
    _check(_bf.bfAffinitySetOpenMPCores(len(cores), arr))
