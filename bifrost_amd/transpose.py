"""bf.transpose (reference python/bifrost/transpose.py surface)."""

import ctypes

from bifrost_amd.libbifrost import _bf, _check
from bifrost_amd.ndarray import asarray

__all__ = ["transpose"]


def transpose(dst, src, axes=None):
    dst_bf = asarray(dst)
    src_bf = asarray(src)
    if axes is None:
        axes = list(reversed(range(len(src_bf.shape))))
    axes_array = (ctypes.c_int * len(axes))(*axes)
    _check(_bf.bfTranspose(src_bf.as_BFarray(), dst_bf.as_BFarray(),
                           axes_array))
    return dst
