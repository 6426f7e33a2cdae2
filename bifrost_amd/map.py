"""bf.map — apply a JIT'd function to named ndarrays (reference
python/bifrost/map.py surface: elementwise AND axis-indexed forms)."""

import ctypes

import numpy as np

from bifrost_amd.libbifrost import _bf, _check, _array
from bifrost_amd.ndarray import asarray, ndarray

__all__ = ["map", "clear_map_cache", "list_map_cache"]


def map(func_string, data, axis_names=None, shape=None, func_name=None,
        extra_code=None, block_shape=None, block_axes=None):
    """Apply `func_string` to the named arrays in `data`.

    Examples::
        bf.map("c = a + b", {'c': c, 'a': a, 'b': b})
        bf.map("a = c.real; b = c.imag", {'c': c, 'a': a, 'b': b})
        bf.map("c = a * s", {'c': c, 'a': a, 's': 2.0})
        bf.map("b(i,j) = a(j,i)", {'a': a, 'b': b},
               axis_names=('i', 'j'), shape=b.shape)
        bf.map("b = a(_ - a.shape()/2)", {'a': a, 'b': b})  # fftshift
    """
    narg = len(data)
    names = []
    arrays = []
    keep_alive = []
    for name, arr in data.items():
        names.append(name)
        if not isinstance(arr, np.ndarray):
            arr = np.asarray(arr)
        if not isinstance(arr, ndarray):
            # scalars / host values: move to device as 0-d-ish arrays
            arr = asarray(np.atleast_1d(arr), space="cuda")
        keep_alive.append(arr)
        arrays.append(arr.as_BFarray())
    ndim = 0
    shape_arr = None
    if shape is not None:
        ndim = len(shape)
        shape_arr = _array(list(shape), dtype=ctypes.c_long)
    axis_arr = _array(list(axis_names)) if axis_names else None
    _check(_bf.bfMap(ndim, shape_arr, axis_arr, narg,
                     _array(arrays), _array(names),
                     (func_name or "").encode() if func_name else None,
                     func_string.encode(),
                     extra_code.encode() if extra_code else None,
                     None, None))


def clear_map_cache():
    _check(_bf.bfMapClearCache())


def list_map_cache():
    """Print bfMap kernel-cache status (reference surface; this backend
    keeps an in-process hipModule cache rather than an on-disk one)."""
    print("Cache enabled: yes (in-process, %d-entry LRU; cleared by "
          "clear_map_cache())" % 128)
