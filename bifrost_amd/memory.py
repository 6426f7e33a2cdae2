"""Raw space-aware allocation helpers (reference python/bifrost/memory.py
surface)."""

import ctypes

from bifrost_amd.libbifrost import _bf, _check, _string2space, _space2string

__all__ = ["raw_malloc", "raw_free", "raw_get_space", "space_accessible",
           "alignment", "memcpy", "memset"]

_ACCESSIBILITY = {
    "system": {"system"},
    "cuda": {"cuda"},
    "cuda_host": {"system", "cuda"},
    "cuda_managed": {"system", "cuda"},
}


def space_accessible(space, from_spaces):
    if isinstance(from_spaces, str):
        from_spaces = [from_spaces]
    space = str(space)
    if space == "auto":
        return True
    acc = _ACCESSIBILITY.get(space, set())
    return any(f in acc for f in from_spaces)


def raw_malloc(size, space):
    ptr = ctypes.c_void_p()
    _check(_bf.bfMalloc(ctypes.byref(ptr), size, _string2space(str(space))))
    return ptr.value


def raw_free(ptr, space="auto"):
    _check(_bf.bfFree(ptr, _string2space(str(space))))


def raw_get_space(ptr):
    space = ctypes.c_int()
    _check(_bf.bfGetSpace(ptr, ctypes.byref(space)))
    return _space2string(space.value)


def alignment():
    return int(_bf.bfGetAlignment())


def memcpy(dst_ptr, dst_space, src_ptr, src_space, count):
    _check(_bf.bfMemcpy(dst_ptr, _string2space(str(dst_space)), src_ptr,
                        _string2space(str(src_space)), count))


def memset(ptr, space, value, count):
    _check(_bf.bfMemset(ptr, _string2space(str(space)), value, count))
