"""FftShiftBlock (reference blocks/fftshift.py surface): cyclic shift
moving the central element to the start, on the bfMap indexed form."""

from copy import deepcopy

import importlib

from bifrost_amd.pipeline import TransformBlock

__all__ = ["FftShiftBlock", "fftshift"]

_map = importlib.import_module("bifrost_amd.map")


class FftShiftBlock(TransformBlock):
    def __init__(self, iring, axes, inverse=False, *args, **kwargs):
        super(FftShiftBlock, self).__init__(iring, *args, **kwargs)
        if not isinstance(axes, (list, tuple)):
            axes = [axes]
        self.specified_axes = axes
        self.inverse = inverse

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        self.axes = [itensor["labels"].index(ax) if isinstance(ax, str)
                     else ax for ax in self.specified_axes]
        frame_axis = itensor["shape"].index(-1)
        if frame_axis in self.axes:
            raise KeyError("Cannot fftshift frame axis")
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]
        oshape = otensor["shape"]
        if "scales" in itensor:
            for ax in self.axes:
                sgn = +1 if self.inverse else -1
                scale_step = otensor["scales"][ax][1]
                otensor["scales"][ax][0] += sgn * (oshape[ax] // 2) * \
                    scale_step
        return ohdr

    def on_data(self, ispan, ospan):
        idata = ispan.data
        odata = ospan.data
        ind_names = ["i%i" % i for i in range(idata.ndim)]
        inds = list(ind_names)
        for ax in self.axes:
            if self.inverse:
                inds[ax] += "-(a.shape(%i)-a.shape(%i)/2)" % (ax, ax)
            else:
                inds[ax] += "-a.shape(%i)/2" % ax
        _map.map("b = a(%s)" % ",".join(inds), shape=idata.shape,
                 axis_names=ind_names, data={"a": idata, "b": odata})


def fftshift(iring, axes, inverse=False, *args, **kwargs):
    """FFT-shift data along the given axes (central element to start).

    Input:  [...], dtype = any, space = CUDA
    Output: [...], same, shifted along `axes`
    """
    return FftShiftBlock(iring, axes, inverse, *args, **kwargs)
