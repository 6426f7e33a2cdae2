"""TransposeBlock (reference blocks/transpose.py surface): permute the
axes of each frame (the time axis may move too)."""

from copy import deepcopy

import bifrost_amd as bf
from bifrost_amd.pipeline import TransformBlock

__all__ = ["TransposeBlock", "transpose"]


class TransposeBlock(TransformBlock):
    def __init__(self, iring, axes, *args, **kwargs):
        super(TransposeBlock, self).__init__(iring, *args, **kwargs)
        self.specified_axes = axes

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        axes = list(self.specified_axes)
        # resolve label names to indices
        for i, axis in enumerate(axes):
            if not isinstance(axis, int):
                axes[i] = itensor["labels"].index(axis)
        self.axes = axes
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]
        for item in ("shape", "labels", "scales", "units"):
            if item in itensor:
                otensor[item] = [itensor[item][a] for a in axes]
        return ohdr

    def on_data(self, ispan, ospan):
        # frame-level axes: prepend the (unmoved) time axis if the time dim
        # is implicit (shape entry -1 at position p in the input tensor).
        idata = ispan.data
        odata = ospan.data
        itime = None
        # axes are tensor-level incl. the -1 time dim; map directly
        bf.transpose(odata, idata, self.axes)


def transpose(iring, axes, *args, **kwargs):
    """Transpose (permute) the axes of the data.

    Input:  [...], dtype = any, space = CUDA
    Output: [axes], dtype = same as input, space = CUDA
    """
    return TransposeBlock(iring, axes, *args, **kwargs)
