"""ctypes wrapper over the reference's own CPU bfUnpack/bfQuantize — ORACLE
VALIDATION ONLY.

Loads oracle/_ref/libbfref_cpu.so (built by oracle/ref_build/Makefile from
the unmodified reference sources with CUDA disabled) and exposes the two ops
on system-space numpy buffers, for cross-validating oracle/bitops.py.
Returns None from load() if the library was never built.
"""

import ctypes
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_HERE, "_ref", "libbfref_cpu.so")

# BFdtype encoding (reference src/bifrost/array.h:43-95)
_INT, _UINT, _FLOAT, _CPLX = 0x0000, 0x0100, 0x0200, 0x100000
DTYPES = {
    "i1": 1 | _INT, "i2": 2 | _INT, "i4": 4 | _INT, "i8": 8 | _INT,
    "i16": 16 | _INT, "i32": 32 | _INT,
    "u2": 2 | _UINT, "u4": 4 | _UINT, "u8": 8 | _UINT,
    "u16": 16 | _UINT, "u32": 32 | _UINT,
    "f32": 32 | _FLOAT, "f64": 64 | _FLOAT,
    "ci1": 1 | _INT | _CPLX, "ci2": 2 | _INT | _CPLX, "ci4": 4 | _INT | _CPLX,
    "ci8": 8 | _INT | _CPLX, "ci16": 16 | _INT | _CPLX, "ci32": 32 | _INT | _CPLX,
    "cf32": 32 | _FLOAT | _CPLX, "cf64": 64 | _FLOAT | _CPLX,
}
_NBIT = {k: (v & 0xFF) * (2 if k.startswith("c") else 1) for k, v in DTYPES.items()}


class _BFarray(ctypes.Structure):
    _fields_ = [("data", ctypes.c_void_p),
                ("space", ctypes.c_int),
                ("dtype", ctypes.c_int),
                ("ndim", ctypes.c_int),
                ("shape", ctypes.c_long * 8),
                ("strides", ctypes.c_long * 8),
                ("immutable", ctypes.c_int),
                ("big_endian", ctypes.c_int),
                ("conjugated", ctypes.c_int)]


def _mk(buf, dtype_name, nelem, big_endian=False, conjugated=False):
    a = _BFarray()
    a.data = buf.ctypes.data
    a.space = 1  # BF_SPACE_SYSTEM
    a.dtype = DTYPES[dtype_name]
    a.ndim = 1
    a.shape[0] = nelem
    a.strides[0] = max(1, _NBIT[dtype_name] // 8)
    a.immutable = 0
    a.big_endian = 1 if big_endian else 0
    a.conjugated = 1 if conjugated else 0
    return a


def load():
    if not os.path.exists(_SO):
        return None
    return ctypes.CDLL(_SO)


def ref_unpack(lib, raw, in_dtype, out_dtype, align_msb=False,
               big_endian=False, conjugate=False):
    """Run the reference CPU bfUnpack on packed bytes; returns raw out bytes."""
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    # _NBIT already folds in the complex factor, so this is the element count.
    nelem = raw.size * 8 // _NBIT[in_dtype]
    out_nbit = _NBIT[out_dtype]
    out = np.zeros(nelem * out_nbit // 8, dtype=np.uint8)
    ia = _mk(raw, in_dtype, nelem, big_endian=big_endian)
    oa = _mk(out, out_dtype, nelem, conjugated=conjugate)
    st = lib.bfUnpack(ctypes.byref(ia), ctypes.byref(oa),
                      ctypes.c_int(1 if align_msb else 0))
    if st != 0:
        raise RuntimeError("reference bfUnpack failed: status %d" % st)
    return out


def ref_quantize(lib, data, out_dtype, scale=1.0):
    """Run the reference CPU bfQuantize on float32 data; returns raw bytes."""
    data = np.ascontiguousarray(data, dtype=np.float32)
    nelem = data.size
    if out_dtype.startswith("c"):
        nelem //= 2
    out_nbit = _NBIT[out_dtype]
    out = np.zeros(max(1, nelem * out_nbit // 8), dtype=np.uint8)
    in_dtype = "cf32" if out_dtype.startswith("c") else "f32"
    ia = _mk(data, in_dtype, nelem)
    oa = _mk(out, out_dtype, nelem)
    lib.bfQuantize.restype = ctypes.c_int
    st = lib.bfQuantize(ctypes.byref(ia), ctypes.byref(oa),
                        ctypes.c_double(scale))
    if st != 0:
        raise RuntimeError("reference bfQuantize failed: status %d" % st)
    return out
