"""bf.unpack (reference python/bifrost/unpack.py surface)."""

from bifrost_amd.libbifrost import _bf, _check
from bifrost_amd.ndarray import asarray

__all__ = ["unpack"]


def unpack(src, dst, align_msb=False):
    src_bf = asarray(src)
    dst_bf = asarray(dst)
    _check(_bf.bfUnpack(src_bf.as_BFarray(), dst_bf.as_BFarray(),
                        1 if align_msb else 0))
    return dst


def unpack_new(src, dst_dtype, align_msb=False):
    src_bf = asarray(src)
    from bifrost_amd.ndarray import ndarray
    dst = ndarray(shape=src_bf.shape, dtype=dst_dtype, space=src_bf.bf.space)
    return unpack(src_bf, dst, align_msb)
