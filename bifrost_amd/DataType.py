"""Bifrost dtype system: string names ('ci8', 'cf32', ...), BFdtype enums
and numpy dtypes.  Source-compatible surface with the reference
python/bifrost/DataType.py (custom struct dtypes for integer complex)."""

import numpy as np

from bifrost_amd.libbifrost import _bf

__all__ = ["DataType", "ci4", "ci8", "ci16", "ci32", "ci64", "cf16"]

# Custom numpy dtypes for integer-complex types (constructed from tuples,
# e.g. np.ndarray([(0,1),(2,3)], dtype=ci8); ci4 packs both nibbles in one
# byte: im low, re high is the *quantize/linalg* convention).
ci4 = np.dtype([("re_im", np.uint8)])
ci8 = np.dtype([("re", np.int8), ("im", np.int8)])
ci16 = np.dtype([("re", np.int16), ("im", np.int16)])
ci32 = np.dtype([("re", np.int32), ("im", np.int32)])
ci64 = np.dtype([("re", np.int64), ("im", np.int64)])
cf16 = np.dtype([("re", np.float16), ("im", np.float16)])

TYPEMAP = {
    "i": {n: getattr(_bf, "BF_DTYPE_I%d" % n) for n in (1, 2, 4, 8, 16, 32, 64)},
    "u": {n: getattr(_bf, "BF_DTYPE_U%d" % n) for n in (1, 2, 4, 8, 16, 32, 64)},
    "f": {n: getattr(_bf, "BF_DTYPE_F%d" % n) for n in (16, 32, 64)},
    "ci": {n: getattr(_bf, "BF_DTYPE_CI%d" % n) for n in (1, 2, 4, 8, 16, 32, 64)},
    "cf": {n: getattr(_bf, "BF_DTYPE_CF%d" % n) for n in (16, 32, 64)},
}
KINDMAP = {
    _bf.BF_DTYPE_INT_TYPE: "i",
    _bf.BF_DTYPE_UINT_TYPE: "u",
    _bf.BF_DTYPE_FLOAT_TYPE: "f",
}
NUMPY_TYPEMAP = {
    "i": {8: np.int8, 16: np.int16, 32: np.int32, 64: np.int64},
    "u": {8: np.uint8, 16: np.uint16, 32: np.uint32, 64: np.uint64},
    "f": {16: np.float16, 32: np.float32, 64: np.float64},
    "ci": {1: np.int8, 2: np.int8, 4: ci4, 8: ci8, 16: ci16, 32: ci32, 64: ci64},
    "cf": {16: cf16, 32: np.complex64, 64: np.complex128},
}

_CI_STRUCTS = (ci4, ci8, ci16, ci32, ci64)


def _is_vector_structure(dt):
    if dt.names is None:
        return False
    names = tuple("f%d" % i for i in range(len(dt.names)))
    return (dt.kind == "V" and dt.names == names and
            all(dt[i] == dt[0] for i in range(1, len(dt.names))))


class DataType(object):
    def __init__(self, t=None):
        if isinstance(t, str):
            i = next(i for i, ch in enumerate(t) if ch.isdigit())
            self._kind = t[:i]
            self._nbit = int(t[i:])
            self._veclen = 1
        elif isinstance(t, int):
            self._nbit = t & _bf.BF_DTYPE_NBIT_BITS
            is_cplx = bool(t & _bf.BF_DTYPE_COMPLEX_BIT)
            self._kind = KINDMAP[t & _bf.BF_DTYPE_TYPE_BITS]
            if is_cplx:
                self._kind = "c" + self._kind
            self._veclen = 1 + ((t & _bf.BF_DTYPE_VECTOR_BITS)
                                >> _bf.BF_DTYPE_VECTOR_BIT0)
        elif isinstance(t, DataType):
            self._kind, self._nbit, self._veclen = t._kind, t._nbit, t._veclen
        elif isinstance(t, tuple):
            self._kind, self._nbit, self._veclen = t
        else:
            t = np.dtype(t)
            ndim = len(t.shape)
            if ndim == 0:
                self._veclen = 1
            elif ndim == 1:
                self._veclen = t.shape[0]
                t = t.base
            else:
                raise TypeError("Unsupported numpy dtype: %r" % (t,))
            self._nbit = t.itemsize * 8
            if _is_vector_structure(t):
                self._veclen = len(t.names)
                t = t[0]
            kind = t.kind
            if kind == "c":
                self._nbit //= 2
                self._kind = "cf"
            elif kind == "V":
                self._nbit //= 2
                if t in _CI_STRUCTS:
                    self._kind = "ci"
                elif t == cf16:
                    self._kind = "cf"
                else:
                    raise TypeError("Unsupported data type: %r" % (t,))
                if t == ci4:
                    self._nbit = 4  # one byte holds both components
            elif kind == "b":
                self._kind = "u"
            elif kind in ("i", "u", "f"):
                self._kind = kind
            else:
                raise TypeError("Unsupported data type: %r" % (t,))

    def __eq__(self, other):
        if not isinstance(other, DataType):
            try:
                other = DataType(other)
            except TypeError:
                return NotImplemented
        return (self._kind == other._kind and self._nbit == other._nbit and
                self._veclen == other._veclen)

    def __ne__(self, other):
        return not (self == other)

    def __hash__(self):
        return hash((self._kind, self._nbit, self._veclen))

    def as_BFdtype(self):
        base = TYPEMAP[self._kind][self._nbit]
        return base | ((self._veclen - 1) << _bf.BF_DTYPE_VECTOR_BIT0)

    def as_numpy_dtype(self):
        base = np.dtype(NUMPY_TYPEMAP[self._kind][self._nbit])
        if self._veclen == 1:
            return base
        return np.dtype(",".join((str(base),) * self._veclen))

    def as_real(self):
        if self.is_complex:
            return DataType((self._kind[1:], self._nbit, self._veclen))
        return self

    def as_complex(self):
        if self.is_complex or self._kind == "u":
            return self
        return DataType(("c" + self._kind, self._nbit, self._veclen))

    def as_floating_point(self):
        """Smallest floating-point type that can represent this type."""
        if self.is_floating_point:
            return self
        nbit = 32 if self._nbit <= 16 else 64
        kind = "cf" if self.is_complex else "f"
        return DataType((kind, nbit, self._veclen))

    def as_integer(self, nbit=None):
        if nbit is None:
            nbit = self._nbit
        kind = "ci" if self.is_complex else "i"
        return DataType((kind, nbit, self._veclen))

    def as_vector(self, veclen):
        return DataType((self._kind, self._nbit, veclen))

    @property
    def is_complex(self):
        return self._kind.startswith("c")

    @property
    def is_real(self):
        return not self.is_complex

    @property
    def is_signed(self):
        return self._kind in ("i", "ci", "f", "cf")

    @property
    def is_floating_point(self):
        return self._kind in ("f", "cf")

    @property
    def is_integer(self):
        return self._kind in ("i", "u", "ci")

    @property
    def itemsize_bits(self):
        return self._nbit * (2 if self.is_complex else 1) * self._veclen

    @property
    def itemsize(self):
        assert self.itemsize_bits % 8 == 0
        return self.itemsize_bits // 8

    def __str__(self):
        if self._veclen == 1:
            return "%s%d" % (self._kind, self._nbit)
        return "%s%d[%d]" % (self._kind, self._nbit, self._veclen)

    def __repr__(self):
        return "DataType('%s')" % str(self)
