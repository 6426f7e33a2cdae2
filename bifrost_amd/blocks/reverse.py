"""ReverseBlock (reference blocks/reverse.py surface): reverse data
along axes via the bfMap negative-index convention a(-i) = a[(n-i)%n]."""

from copy import deepcopy

import importlib

from bifrost_amd.pipeline import TransformBlock

__all__ = ["ReverseBlock", "reverse"]

_map = importlib.import_module("bifrost_amd.map")


class ReverseBlock(TransformBlock):
    def __init__(self, iring, axes, *args, **kwargs):
        super(ReverseBlock, self).__init__(iring, *args, **kwargs)
        if not isinstance(axes, (list, tuple)):
            axes = [axes]
        self.specified_axes = axes

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        self.axes = [itensor["labels"].index(ax) if isinstance(ax, str)
                     else ax for ax in self.specified_axes]
        frame_axis = itensor["shape"].index(-1)
        if frame_axis in self.axes:
            raise KeyError("Cannot reverse frame axis")
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]
        oshape = otensor["shape"]
        if "scales" in itensor:
            for ax in self.axes:
                scale_step = otensor["scales"][ax][1]
                otensor["scales"][ax][0] += oshape[ax] * scale_step
                otensor["scales"][ax][1] = -scale_step
        return ohdr

    def on_data(self, ispan, ospan):
        idata = ispan.data
        odata = ospan.data
        ind_names = ["i%i" % i for i in range(idata.ndim)]
        inds = list(ind_names)
        for ax in self.axes:
            inds[ax] = "-" + inds[ax]
        _map.map("b = a(%s)" % ",".join(inds), shape=idata.shape,
                 axis_names=ind_names, data={"a": idata, "b": odata})


def reverse(iring, axes, *args, **kwargs):
    """Reverse data along the given axes (a[-i] convention: element 0
    stays put, the rest reverse — the frequency-reversal convention).

    Input:  [...], dtype = any, space = CUDA
    Output: [...], same, reversed along `axes`
    """
    return ReverseBlock(iring, axes, *args, **kwargs)
