"""ctypes FFI over libbifrost.so (the MI355X-native C ABI).

Hand-written equivalent of the reference's ctypesgen-generated module
(python/bifrost/libbifrost.py + libbifrost_generated.py): the `_bf`
namespace carries the constants, structs and function prototypes; `_check`
maps BFstatus to exceptions.  The library MUST be present and loadable —
there is deliberately no fallback: ops fail loudly if the HIP extension is
missing.
"""

import ctypes
import os

__all__ = ["_bf", "_check", "_get", "_array", "BifrostObject", "EndOfDataStop",
           "_string2space", "_space2string"]

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "lib", "libbifrost.so")


class _BFNamespace(object):
    """Namespace mirroring the generated bindings module."""
    pass


_bf = _BFNamespace()

# ---- constants (ABI contract, include/bifrost/*.h) ------------------------
_bf.BF_STATUS_SUCCESS = 0
_bf.BF_STATUS_END_OF_DATA = 1
_bf.BF_STATUS_WOULD_BLOCK = 2

_bf.BF_MAX_DIMS = 8

_bf.BF_DTYPE_NBIT_BITS = 0x0000FF
_bf.BF_DTYPE_TYPE_BITS = 0x000F00
_bf.BF_DTYPE_VECTOR_BITS = 0x0FF000
_bf.BF_DTYPE_VECTOR_BIT0 = 12
_bf.BF_DTYPE_COMPLEX_BIT = 0x100000
_bf.BF_DTYPE_INT_TYPE = 0x0000
_bf.BF_DTYPE_UINT_TYPE = 0x0100
_bf.BF_DTYPE_FLOAT_TYPE = 0x0200
_bf.BF_DTYPE_STRING_TYPE = 0x0300
_bf.BF_DTYPE_STORAGE_TYPE = 0x0400

for _name, _base in (("I", _bf.BF_DTYPE_INT_TYPE), ("U", _bf.BF_DTYPE_UINT_TYPE)):
    for _nbit in (1, 2, 4, 8, 16, 32, 64):
        setattr(_bf, "BF_DTYPE_%s%d" % (_name, _nbit), _nbit | _base)
for _nbit in (16, 32, 64):
    setattr(_bf, "BF_DTYPE_F%d" % _nbit, _nbit | _bf.BF_DTYPE_FLOAT_TYPE)
for _nbit in (1, 2, 4, 8, 16, 32, 64):
    setattr(_bf, "BF_DTYPE_CI%d" % _nbit,
            _nbit | _bf.BF_DTYPE_INT_TYPE | _bf.BF_DTYPE_COMPLEX_BIT)
for _nbit in (16, 32, 64):
    setattr(_bf, "BF_DTYPE_CF%d" % _nbit,
            _nbit | _bf.BF_DTYPE_FLOAT_TYPE | _bf.BF_DTYPE_COMPLEX_BIT)

_bf.BF_SPACE_AUTO = 0
_bf.BF_SPACE_SYSTEM = 1
_bf.BF_SPACE_CUDA = 2
_bf.BF_SPACE_CUDA_HOST = 3
_bf.BF_SPACE_CUDA_MANAGED = 4

# Config flags the reference exposes through libbifrost_generated
_bf.BF_CUDA_ENABLED = 1      # HIP backend: device support is present
_bf.BF_FLOAT128_ENABLED = 0
_bf.BF_DEBUG_ENABLED = 0
_bf.BF_TRACE_ENABLED = 0

BFstatus = ctypes.c_int
BFbool = ctypes.c_int
BFsize = ctypes.c_ulong
BFoffset = ctypes.c_ulonglong
BFdtype = ctypes.c_int
BFspace = ctypes.c_int

_bf.BFstatus = BFstatus
_bf.BFbool = BFbool
_bf.BFsize = BFsize
_bf.BFoffset = BFoffset
_bf.BFdtype = BFdtype
_bf.BFspace = BFspace


class BFarray(ctypes.Structure):
    _fields_ = [("data", ctypes.c_void_p),
                ("space", ctypes.c_int),
                ("dtype", ctypes.c_int),
                ("ndim", ctypes.c_int),
                ("shape", ctypes.c_long * 8),
                ("strides", ctypes.c_long * 8),
                ("immutable", ctypes.c_int),
                ("big_endian", ctypes.c_int),
                ("conjugated", ctypes.c_int)]


class BFsequence_info(ctypes.Structure):
    _fields_ = [("ring", ctypes.c_void_p),
                ("name", ctypes.c_char_p),
                ("time_tag", BFoffset),
                ("header", ctypes.c_void_p),
                ("header_size", BFsize),
                ("nringlet", BFsize)]


class BFspan_info(ctypes.Structure):
    _fields_ = [("ring", ctypes.c_void_p),
                ("data", ctypes.c_void_p),
                ("size", BFsize),
                ("stride", BFsize),
                ("offset", BFsize),
                ("nringlet", BFsize)]


_bf.BFarray = BFarray
_bf.struct_BFarray_ = BFarray
_bf.BFsequence_info = BFsequence_info
_bf.BFspan_info = BFspan_info

# Opaque handles
for _h in ("BFlinalg", "BFring", "BFsequence", "BFrsequence", "BFwsequence",
           "BFspan", "BFrspan", "BFwspan", "BFproclog", "BFfft"):
    setattr(_bf, _h, type(_h, (ctypes.c_void_p,), {}))

_lib = ctypes.CDLL(_LIB_PATH, mode=ctypes.RTLD_GLOBAL)

_PA = ctypes.POINTER(BFarray)


def _proto(name, restype, *argtypes):
    fn = getattr(_lib, name)
    fn.restype = restype
    fn.argtypes = list(argtypes)
    setattr(_bf, name, fn)
    return fn


c_int_p = ctypes.POINTER(ctypes.c_int)
c_void_pp = ctypes.POINTER(ctypes.c_void_p)
c_char_pp = ctypes.POINTER(ctypes.c_char_p)
size_p = ctypes.POINTER(BFsize)
off_p = ctypes.POINTER(BFoffset)

# common / cuda
_proto("bfGetStatusString", ctypes.c_char_p, BFstatus)
_proto("bfGetDebugEnabled", BFbool)
_proto("bfSetDebugEnabled", BFstatus, BFbool)
_proto("bfGetCudaEnabled", BFbool)
_proto("bfStreamGet", BFstatus, ctypes.c_void_p)
_proto("bfStreamSet", BFstatus, ctypes.c_void_p)
_proto("bfStreamSynchronize", BFstatus)
_proto("bfDeviceGet", BFstatus, c_int_p)
_proto("bfDeviceSet", BFstatus, ctypes.c_int)
_proto("bfDeviceSetById", BFstatus, ctypes.c_char_p)
_proto("bfDevicesSetNoSpinCPU", BFstatus)

# memory
_proto("bfMalloc", BFstatus, c_void_pp, BFsize, BFspace)
_proto("bfFree", BFstatus, ctypes.c_void_p, BFspace)
_proto("bfGetSpace", BFstatus, ctypes.c_void_p, c_int_p)
_proto("bfGetSpaceString", ctypes.c_char_p, BFspace)
_proto("bfMemcpy", BFstatus, ctypes.c_void_p, BFspace, ctypes.c_void_p,
       BFspace, BFsize)
_proto("bfMemcpy2D", BFstatus, ctypes.c_void_p, BFsize, BFspace,
       ctypes.c_void_p, BFsize, BFspace, BFsize, BFsize)
_proto("bfMemset", BFstatus, ctypes.c_void_p, BFspace, ctypes.c_int, BFsize)
_proto("bfMemset2D", BFstatus, ctypes.c_void_p, BFsize, BFspace, ctypes.c_int,
       BFsize, BFsize)
_proto("bfGetAlignment", BFsize)

# array
_proto("bfArrayMalloc", BFstatus, _PA)
_proto("bfArrayFree", BFstatus, _PA)
_proto("bfArrayCopy", BFstatus, _PA, _PA)
_proto("bfArrayMemset", BFstatus, _PA, ctypes.c_int)

# ops
_proto("bfLinAlgCreate", BFstatus, ctypes.POINTER(_bf.BFlinalg))
_proto("bfLinAlgDestroy", BFstatus, _bf.BFlinalg)
_proto("bfLinAlgMatMul", BFstatus, _bf.BFlinalg, ctypes.c_double, _PA, _PA,
       ctypes.c_double, _PA)
_proto("bfTranspose", BFstatus, _PA, _PA, c_int_p)
_proto("bfUnpack", BFstatus, _PA, _PA, BFbool)
_proto("bfQuantize", BFstatus, _PA, _PA, ctypes.c_double)
_proto("bfMap", BFstatus, ctypes.c_int, ctypes.POINTER(ctypes.c_long),
       ctypes.POINTER(ctypes.c_char_p), ctypes.c_int,
       ctypes.POINTER(ctypes.POINTER(BFarray)),
       ctypes.POINTER(ctypes.c_char_p), ctypes.c_char_p, ctypes.c_char_p,
       ctypes.c_char_p, c_int_p, c_int_p)
_proto("bfMapClearCache", BFstatus)
_proto("bfReduce", BFstatus, _PA, _PA, ctypes.c_int)
_proto("bfFftCreate", BFstatus, ctypes.POINTER(_bf.BFfft))
_proto("bfFftDestroy", BFstatus, _bf.BFfft)
_proto("bfFftInit", BFstatus, _bf.BFfft, _PA, _PA, ctypes.c_int, c_int_p,
       BFbool, ctypes.POINTER(ctypes.c_size_t))
_proto("bfFftExecute", BFstatus, _bf.BFfft, _PA, _PA, BFbool,
       ctypes.c_void_p, ctypes.c_size_t)

# proclog / affinity
_proto("bfProcLogCreate", BFstatus, ctypes.POINTER(_bf.BFproclog),
       ctypes.c_char_p)
_proto("bfProcLogDestroy", BFstatus, _bf.BFproclog)
_proto("bfProcLogUpdate", BFstatus, _bf.BFproclog, ctypes.c_char_p)
_proto("bfAffinitySetCore", BFstatus, ctypes.c_int)
_proto("bfAffinityGetCore", BFstatus, c_int_p)
_proto("bfAffinitySetOpenMPCores", BFstatus, BFsize, c_int_p)

# ring
_proto("bfTestSuite", ctypes.c_int)
_proto("bfRingCreate", BFstatus, ctypes.POINTER(_bf.BFring), ctypes.c_char_p,
       BFspace)
_proto("bfRingDestroy", BFstatus, _bf.BFring)
_proto("bfRingResize", BFstatus, _bf.BFring, BFsize, BFsize, BFsize)
_proto("bfRingGetName", BFstatus, _bf.BFring, c_char_pp)
_proto("bfRingGetSpace", BFstatus, _bf.BFring, c_int_p)
_proto("bfRingSetAffinity", BFstatus, _bf.BFring, ctypes.c_int)
_proto("bfRingGetAffinity", BFstatus, _bf.BFring, c_int_p)
_proto("bfRingLock", BFstatus, _bf.BFring)
_proto("bfRingUnlock", BFstatus, _bf.BFring)
_proto("bfRingLockedGetData", BFstatus, _bf.BFring, c_void_pp)
_proto("bfRingLockedGetContiguousSpan", BFstatus, _bf.BFring, size_p)
_proto("bfRingLockedGetTotalSpan", BFstatus, _bf.BFring, size_p)
_proto("bfRingLockedGetNRinglet", BFstatus, _bf.BFring, size_p)
_proto("bfRingLockedGetStride", BFstatus, _bf.BFring, size_p)
_proto("bfRingBeginWriting", BFstatus, _bf.BFring)
_proto("bfRingEndWriting", BFstatus, _bf.BFring)
_proto("bfRingWritingEnded", BFstatus, _bf.BFring, c_int_p)
_proto("bfRingSequenceBegin", BFstatus, ctypes.POINTER(_bf.BFwsequence),
       _bf.BFring, ctypes.c_char_p, BFoffset, BFsize, ctypes.c_void_p,
       BFsize, BFoffset)
_proto("bfRingSequenceEnd", BFstatus, _bf.BFwsequence, BFoffset)
_proto("bfRingSequenceOpen", BFstatus, ctypes.POINTER(_bf.BFrsequence),
       _bf.BFring, ctypes.c_char_p, BFbool)
_proto("bfRingSequenceOpenAt", BFstatus, ctypes.POINTER(_bf.BFrsequence),
       _bf.BFring, BFoffset, BFbool)
_proto("bfRingSequenceOpenLatest", BFstatus, ctypes.POINTER(_bf.BFrsequence),
       _bf.BFring, BFbool)
_proto("bfRingSequenceOpenEarliest", BFstatus, ctypes.POINTER(_bf.BFrsequence),
       _bf.BFring, BFbool)
_proto("bfRingSequenceNext", BFstatus, _bf.BFrsequence)
_proto("bfRingSequenceClose", BFstatus, _bf.BFrsequence)
_proto("bfRingSequenceGetRing", BFstatus, _bf.BFsequence,
       ctypes.POINTER(_bf.BFring))
_proto("bfRingSequenceGetName", BFstatus, _bf.BFsequence, c_char_pp)
_proto("bfRingSequenceGetTimeTag", BFstatus, _bf.BFsequence, off_p)
_proto("bfRingSequenceGetHeader", BFstatus, _bf.BFsequence, c_void_pp)
_proto("bfRingSequenceGetHeaderSize", BFstatus, _bf.BFsequence, size_p)
_proto("bfRingSequenceGetNRinglet", BFstatus, _bf.BFsequence, size_p)
_proto("bfRingSequenceGetInfo", BFstatus, _bf.BFsequence,
       ctypes.POINTER(BFsequence_info))
_proto("bfRingSpanReserve", BFstatus, ctypes.POINTER(_bf.BFwspan), _bf.BFring,
       BFsize, BFbool)
_proto("bfRingSpanCommit", BFstatus, _bf.BFwspan, BFsize)
_proto("bfRingSpanAcquire", BFstatus, ctypes.POINTER(_bf.BFrspan),
       _bf.BFrsequence, BFoffset, BFsize)
_proto("bfRingSpanRelease", BFstatus, _bf.BFrspan)
_proto("bfRingSpanGetSizeOverwritten", BFstatus, _bf.BFrspan, size_p)
_proto("bfRingSpanGetRing", BFstatus, _bf.BFspan, ctypes.POINTER(_bf.BFring))
_proto("bfRingSpanGetData", BFstatus, _bf.BFspan, c_void_pp)
_proto("bfRingSpanGetSize", BFstatus, _bf.BFspan, size_p)
_proto("bfRingSpanGetStride", BFstatus, _bf.BFspan, size_p)
_proto("bfRingSpanGetOffset", BFstatus, _bf.BFspan, size_p)
_proto("bfRingSpanGetNRinglet", BFstatus, _bf.BFspan, size_p)
_proto("bfRingSpanGetInfo", BFstatus, _bf.BFspan,
       ctypes.POINTER(BFspan_info))


class EndOfDataStop(RuntimeError):
    """Raised for BF_STATUS_END_OF_DATA (PEP479-safe StopIteration stand-in)."""
    pass


def _check(status):
    if status == _bf.BF_STATUS_SUCCESS:
        return status
    if status == _bf.BF_STATUS_END_OF_DATA:
        raise EndOfDataStop("BF_STATUS_END_OF_DATA")
    if status == _bf.BF_STATUS_WOULD_BLOCK:
        raise IOError("BF_STATUS_WOULD_BLOCK")
    raise RuntimeError(_bf.bfGetStatusString(status).decode())


def _get(func, *args):
    """Call func(*args, &out) where out's type comes from the last argtype."""
    out_t = func.argtypes[-1]._type_
    out = out_t()
    _check(func(*(args + (ctypes.byref(out),))))
    return out.value


def _array(size_or_vals, dtype=None):
    if size_or_vals is None:
        return None
    try:
        iter(size_or_vals)
    except TypeError:
        return (dtype * size_or_vals)()
    vals = list(size_or_vals)
    if not vals:
        return None
    if dtype is None:
        if isinstance(vals[0], int):
            dtype = ctypes.c_int
        elif isinstance(vals[0], float):
            dtype = ctypes.c_double
        elif isinstance(vals[0], str):
            vals = [v.encode() for v in vals]
            dtype = ctypes.c_char_p
        elif isinstance(vals[0], BFarray):
            dtype = ctypes.POINTER(BFarray)
            vals = [ctypes.pointer(v) for v in vals]
        else:
            raise TypeError("Cannot deduce C type from %r" % type(vals[0]))
    return (dtype * len(vals))(*vals)


_SPACEMAP = {"auto": _bf.BF_SPACE_AUTO, "system": _bf.BF_SPACE_SYSTEM,
             "cuda": _bf.BF_SPACE_CUDA, "cuda_host": _bf.BF_SPACE_CUDA_HOST,
             "cuda_managed": _bf.BF_SPACE_CUDA_MANAGED}
_SPACEMAP_INV = {v: k for k, v in _SPACEMAP.items()}


def _string2space(s):
    try:
        return _SPACEMAP[s]
    except KeyError:
        raise ValueError("Invalid space: %r" % (s,))


def _space2string(i):
    return _SPACEMAP_INV[int(i)]


class BifrostObject(object):
    """Base for simple objects with Create/Destroy C functions."""

    def __init__(self, constructor, destructor, *args):
        self.obj = destructor.argtypes[0]()
        _check(constructor(ctypes.byref(self.obj), *args))
        self._destructor = destructor

    def _destroy(self):
        if getattr(self, "obj", None):
            _check(self._destructor(self.obj))
            self.obj.value = 0

    def __del__(self):
        try:
            self._destroy()
        except Exception:
            pass

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self._destroy()
