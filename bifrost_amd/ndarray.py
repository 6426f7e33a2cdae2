"""Space-aware ndarray: np.ndarray subclass carrying bifrost metadata
(space, bifrost dtype, byte order, conjugated flag).

Source-compatible surface with the reference python/bifrost/ndarray.py:
device arrays wrap HIP pointers in a numpy view (metadata ops — reshape,
transpose, slicing — work; host math on device memory is invalid, same as
the reference).  Sub-byte dtypes (ci4) fold their packing into the last
dim: numpy shape counts BYTES, as_BFarray scales the last dim back to
elements (reference ndarray.py:239-247,329-332 semantics).
"""

import ctypes

import numpy as np

from bifrost_amd import device
from bifrost_amd.DataType import DataType
from bifrost_amd.Space import Space
from bifrost_amd.libbifrost import _bf, _check
from bifrost_amd.memory import (raw_free, raw_get_space, raw_malloc,
                                space_accessible)

__all__ = ["ndarray", "asarray", "empty", "empty_like", "zeros", "zeros_like",
           "copy_array", "memset_array"]


def _address_as_buffer(address, nbyte, readonly=False):
    if address is None:
        raise ValueError("Cannot create buffer from NULL pointer")
    fn = ctypes.pythonapi.PyMemoryView_FromMemory
    fn.restype = ctypes.py_object
    fn.argtypes = (ctypes.c_void_p, ctypes.c_ssize_t, ctypes.c_int)
    return fn(address, nbyte, 0x100 if readonly else 0x200)


def asarray(arr, space=None):
    if isinstance(arr, ndarray) and (space is None or space == arr.bf.space):
        return arr
    return ndarray(arr, space=space)


def empty(shape, dtype="f32", space=None, **kwargs):
    return ndarray(shape=shape, dtype=dtype, space=space, **kwargs)


def empty_like(arr, space=None):
    arr = asarray(arr)
    if space is None:
        space = arr.bf.space
    return ndarray(shape=arr.shape, dtype=arr.bf.dtype, space=space,
                   native=arr.bf.native, conjugated=arr.bf.conjugated)


def zeros(shape, dtype="f32", space=None, **kwargs):
    ret = empty(shape, dtype, space, **kwargs)
    memset_array(ret, 0)
    return ret


def zeros_like(arr, space=None):
    ret = empty_like(arr, space)
    memset_array(ret, 0)
    return ret


def copy_array(dst, src):
    dst_bf = asarray(dst)
    src_bf = asarray(src)
    if (space_accessible(dst_bf.bf.space, ["system"]) and
            space_accessible(src_bf.bf.space, ["system"])):
        if "cuda_managed" in (dst_bf.bf.space, src_bf.bf.space):
            device.stream_synchronize()
        np.copyto(dst_bf, src_bf)
    else:
        _check(_bf.bfArrayCopy(dst_bf.as_BFarray(), src_bf.as_BFarray()))
        if dst_bf.bf.space != src_bf.bf.space:
            device.stream_synchronize()
    return dst


def memset_array(dst, value):
    dst = asarray(dst)
    _check(_bf.bfArrayMemset(dst.as_BFarray(), value))
    return dst


class BFArrayInfo(object):
    def __init__(self, space, dtype, native, conjugated, ownbuffer=None):
        self.space = space
        self.dtype = dtype
        self.native = native
        self.conjugated = conjugated
        self.ownbuffer = ownbuffer


class ndarray(np.ndarray):
    def __new__(cls, base=None, space=None, shape=None, dtype=None,
                buffer=None, offset=0, strides=None, native=None,
                conjugated=None):
        if isinstance(shape, int):
            shape = [shape]
        if base is not None:
            if (shape is not None or buffer is not None or offset != 0 or
                    strides is not None or native is not None):
                raise ValueError("Invalid argument combination with base")
            if isinstance(base, _bf.BFarray):
                ndim = base.ndim
                return ndarray.__new__(
                    cls, space=Space(base.space)._space,
                    buffer=int(base.data),
                    shape=list(base.shape)[:ndim],
                    dtype=DataType(int(base.dtype)),
                    strides=list(base.strides)[:ndim])
            if dtype is not None:
                dtype = DataType(dtype)
            if space is None and dtype is None:
                if not isinstance(base, np.ndarray):
                    base = np.asarray(base)
                obj = base.view(cls)
                if conjugated is not None:
                    obj.bf.conjugated = conjugated
                return obj
            # Copy/convert path
            if not isinstance(base, np.ndarray):
                if dtype is not None:
                    base = np.array(base, dtype=dtype.as_numpy_dtype())
                else:
                    base = np.array(base)
            if not isinstance(base, ndarray) and dtype is not None:
                base = base.astype(dtype.as_numpy_dtype())
            base = ndarray(base)  # view as bifrost ndarray
            if dtype is not None and base.bf.dtype != dtype:
                raise TypeError("Cannot convert %s to %s during construction"
                                % (base.bf.dtype, dtype))
            if conjugated is None:
                conjugated = base.bf.conjugated
            obj = ndarray.__new__(cls, space=space, shape=base.shape,
                                  dtype=base.bf.dtype, strides=base.strides,
                                  native=base.bf.native,
                                  conjugated=conjugated)
            copy_array(obj, base)
            return obj

        # base is None: allocate or wrap
        if dtype is None:
            dtype = "f32"
        dtype = DataType(dtype)
        if native is None:
            native = True
        if conjugated is None:
            conjugated = False
        ownbuffer = None
        if strides is None:
            itemsize_bits = dtype.itemsize_bits
            if itemsize_bits < 8:
                pack_factor = 8 // itemsize_bits
                if not len(shape) or shape[-1] % pack_factor != 0:
                    raise ValueError("Array cannot be packed")
                shape = list(shape)
                shape[-1] //= pack_factor
                itemsize = 1
            else:
                itemsize = itemsize_bits // 8
            if len(shape):
                strides = [itemsize]
                for dim in reversed(shape[1:]):
                    strides.append(strides[-1] * dim)
                strides = tuple(reversed(strides))
            else:
                strides = tuple()
        nbyte = strides[0] * shape[0] if len(shape) else dtype.itemsize
        if buffer is None:
            if space is None:
                space = "system"
            ownbuffer = raw_malloc(nbyte, space)
            buffer = ownbuffer
        else:
            if space is None:
                space = raw_get_space(buffer)
        space = str(Space(space))
        dtype_np = np.dtype(dtype.as_numpy_dtype())
        if not native:
            dtype_np = dtype_np.newbyteorder()
        data_buffer = _address_as_buffer(buffer, nbyte)
        obj = np.ndarray.__new__(cls, shape, dtype_np, data_buffer, offset,
                                 strides)
        obj.bf = BFArrayInfo(space, dtype, native, conjugated, ownbuffer)
        return obj

    def __array_finalize__(self, obj):
        if obj is None:
            return
        if isinstance(obj, ndarray) and hasattr(obj, "bf"):
            self.bf = BFArrayInfo(obj.bf.space, obj.bf.dtype, obj.bf.native,
                                  obj.bf.conjugated)
        else:
            self.bf = BFArrayInfo("system", DataType(obj.dtype),
                                  obj.dtype.isnative, False)

    def __del__(self):
        if hasattr(self, "bf") and self.bf.ownbuffer:
            try:
                raw_free(self.bf.ownbuffer, self.bf.space)
            except (AttributeError, TypeError):
                pass  # interpreter shutdown: module globals already torn down

    def as_BFarray(self):
        a = _bf.BFarray()
        a.data = self.ctypes.data
        a.space = Space(self.bf.space).as_BFspace()
        a.dtype = self.bf.dtype.as_BFdtype()
        a.immutable = not self.flags["WRITEABLE"]
        a.ndim = len(self.shape)
        if a.ndim == 0:
            a.ndim = 1
            a.shape[0] = 1
            a.strides[0] = self.bf.dtype.itemsize
            a.big_endian = not self.bf.native
            a.conjugated = self.bf.conjugated
            return a
        for d in range(len(self.shape)):
            a.shape[d] = self.shape[d]
        itemsize_bits = self.bf.dtype.itemsize_bits
        if itemsize_bits < 8:
            a.shape[a.ndim - 1] *= 8 // itemsize_bits
        for d in range(len(self.strides)):
            a.strides[d] = self.strides[d]
        a.big_endian = not self.bf.native
        a.conjugated = self.bf.conjugated
        return a

    def conj(self):
        return ndarray(self, conjugated=not self.bf.conjugated)

    def view(self, dtype=None, type_=None):
        if type_ is not None:
            dtype = type_
        if isinstance(dtype, type) and issubclass(dtype, np.ndarray):
            return super(ndarray, self).view(dtype)
        dtype_bf = DataType(dtype)
        dtype_np = np.dtype(dtype_bf.as_numpy_dtype())
        v = super(ndarray, self).view(dtype_np)
        v.bf.dtype = dtype_bf
        return v

    def astype(self, dtype):
        dtype_bf = DataType(dtype)
        if space_accessible(self.bf.space, ["system"]):
            if self.bf.space == "cuda_managed":
                device.stream_synchronize()
            if dtype_bf.is_complex and dtype_bf.is_integer:
                a = ndarray(shape=self.shape, dtype=dtype_bf)
                a["re"] = self.real.astype(dtype_bf.as_real().as_numpy_dtype())
                a["im"] = self.imag.astype(dtype_bf.as_real().as_numpy_dtype())
            else:
                a = super(ndarray, self).astype(dtype_bf.as_numpy_dtype())
                a = asarray(a)
            a.bf.dtype = dtype_bf
            return a
        # device arrays: packed ci4 goes through the native unpack/
        # quantize kernels (bfMap cannot address sub-byte elements) ...
        if self.bf.dtype == DataType("ci4") or dtype_bf == DataType("ci4"):
            # (package attrs 'unpack'/'quantize' are rebound to functions
            # by __init__, so fetch the real modules)
            import importlib
            _quantize = importlib.import_module("bifrost_amd.quantize")
            _unpack = importlib.import_module("bifrost_amd.unpack")
            a = ndarray(shape=self.shape, dtype=dtype_bf,
                        space=self.bf.space)
            if self.bf.dtype == DataType("ci4"):
                if dtype_bf in (DataType("ci8"), DataType("cf32"),
                                DataType("cf64")):
                    _unpack.unpack(self, a)
                    return a
                tmp = self.astype("cf32")
                return tmp.astype(dtype)
            _quantize.quantize(self, a, scale=1.0)
            return a
        # ... everything else JITs an elementwise conversion through
        # bfMap.  The package attribute 'map' is rebound to the function
        # by __init__, so fetch the real module from sys.modules.
        import importlib
        bf_map = importlib.import_module("bifrost_amd.map")
        a = ndarray(shape=self.shape, dtype=dtype_bf, space=self.bf.space)
        if dtype_bf.is_complex:
            if self.bf.dtype.is_complex:
                func = "a.real = b.real; a.imag = b.imag"
            else:
                func = "a.real = b; a.imag = 0"
        else:
            if self.bf.dtype.is_complex:
                func = "a = b.real"
            else:
                func = "a = b"
        bf_map.map(func, {"a": a, "b": self})
        return a

    def _system_accessible_copy(self):
        if space_accessible(self.bf.space, ["system"]):
            return self
        return self.copy(space="system")

    def tofile(self, fid, sep="", format="%s"):
        # device arrays stage through a system copy first (reference
        # ndarray.py tofile override)
        return super(ndarray,
                     self._system_accessible_copy()).tofile(fid, sep, format)

    def __repr__(self):
        return super(ndarray, self._system_accessible_copy()).__repr__()

    def __str__(self):
        return super(ndarray, self._system_accessible_copy()).__str__()

    def byteswap(self, inplace=False):
        if inplace:
            self.bf.native = not self.bf.native
            return super(ndarray, self).byteswap(True)
        return ndarray(self).byteswap(True)

    def copy(self, space=None, order="C"):
        if order != "C":
            raise NotImplementedError("Only order='C' is supported")
        if space is None:
            space = self.bf.space
        if not self.flags["C_CONTIGUOUS"]:
            if space_accessible(self.bf.space, ["system"]):
                if space == "cuda_managed":
                    device.stream_synchronize()
                temp = ndarray(shape=self.shape, dtype=self.bf.dtype,
                               space=self.bf.space)
                temp[...] = np.array(self).copy()
                if self.bf.space != space:
                    return ndarray(temp, space=space)
                return temp
            # Device: materialize via bfTranspose from the stride-sorted view
            permute = list(np.argsort(self.strides)[::-1])
            self_corder = self.as_BFarray()
            shape_type = ctypes.c_long * _bf.BF_MAX_DIMS
            c_shape = [self.shape[p] for p in permute]
            c_strides = [self.strides[p] for p in permute]
            # NOTE: packed last-dim scaling only applies to contiguous views
            self_corder.shape = shape_type(*(c_shape + [0] * (8 - len(c_shape))))
            self_corder.strides = shape_type(*(c_strides + [0] * (8 - len(c_strides))))
            temp = ndarray(shape=self.shape, dtype=self.bf.dtype,
                           space=self.bf.space)
            axes_array = (ctypes.c_int * self.ndim)(*permute)
            _check(_bf.bfTranspose(self_corder, temp.as_BFarray(), axes_array))
            if self.bf.space != space:
                return ndarray(temp, space=space)
            return temp
        return ndarray(self, space=space)

    def _key_returns_scalar(self, key):
        if isinstance(key, tuple):
            if len(key) == len(self.shape):
                if all(not isinstance(k, slice) for k in key):
                    return True
        elif not isinstance(key, slice):
            if self.ndim <= 1:
                return True
        return False

    def __getitem__(self, key):
        if (self._key_returns_scalar(key) and
                not space_accessible(self.bf.space, ["system"])):
            return super(ndarray,
                         self._system_accessible_copy()).__getitem__(key)
        return super(ndarray, self).__getitem__(key)

    def __setitem__(self, key, val):
        if space_accessible(self.bf.space, ["system"]):
            return super(ndarray, self).__setitem__(key, val)
        if self._key_returns_scalar(key):
            if isinstance(key, tuple):
                key = (slice(key[0], key[0] + 1),) + key[1:]
            else:
                key = slice(key, key + 1)
        copy_array(super(ndarray, self).__getitem__(key), val)
