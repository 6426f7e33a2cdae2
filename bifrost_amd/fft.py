"""Fft: FFT plans over ndarray axes (reference python/bifrost/fft.py
surface; hipFFT backend, unnormalized cuFFT convention)."""

import ctypes

from bifrost_amd.libbifrost import _bf, _check, BifrostObject
from bifrost_amd.ndarray import asarray

__all__ = ["Fft"]


class Fft(BifrostObject):
    def __init__(self):
        BifrostObject.__init__(self, _bf.bfFftCreate, _bf.bfFftDestroy)

    def init(self, iarray, oarray, axes=None, apply_fftshift=False):
        if axes is None:
            axes = [iarray.ndim - 1]
        elif isinstance(axes, int):
            axes = [axes]
        axes = [a + iarray.ndim if a < 0 else a for a in axes]
        axes_arr = (ctypes.c_int * len(axes))(*axes)
        size = ctypes.c_size_t()
        _check(_bf.bfFftInit(self.obj, asarray(iarray).as_BFarray(),
                             asarray(oarray).as_BFarray(), len(axes),
                             axes_arr, apply_fftshift, ctypes.byref(size)))
        return size.value

    def execute(self, iarray, oarray, inverse=False):
        _check(_bf.bfFftExecute(self.obj, asarray(iarray).as_BFarray(),
                                asarray(oarray).as_BFarray(), inverse,
                                None, 0))
        return oarray
