"""bf.quantize (reference python/bifrost/quantize.py surface)."""

from bifrost_amd.libbifrost import _bf, _check
from bifrost_amd.ndarray import asarray

__all__ = ["quantize"]


def quantize(src, dst, scale=1.0):
    src_bf = asarray(src)
    dst_bf = asarray(dst)
    _check(_bf.bfQuantize(src_bf.as_BFarray(), dst_bf.as_BFarray(),
                          float(scale)))
    return dst
