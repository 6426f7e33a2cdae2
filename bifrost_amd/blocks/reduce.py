"""ReduceBlock (reference blocks/reduce.py surface): reduce an axis by a
factor with sum/mean/min/max/stderr or their power (|x|^2) variants."""

from copy import deepcopy

from bifrost_amd.pipeline import TransformBlock
from bifrost_amd.reduce import reduce as bf_reduce

__all__ = ["ReduceBlock", "reduce"]


class ReduceBlock(TransformBlock):
    def __init__(self, iring, axis, factor=None, op="sum", *args, **kwargs):
        super(ReduceBlock, self).__init__(iring, *args, **kwargs)
        self.specified_axis = axis
        self.specified_factor = factor
        self.op = op

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]
        otensor["dtype"] = "f32"
        if itensor["dtype"] == "cf32" and not self.op.startswith("pwr"):
            otensor["dtype"] = "cf32"
        if "labels" in itensor and isinstance(self.specified_axis, str):
            self.axis = itensor["labels"].index(self.specified_axis)
        else:
            self.axis = self.specified_axis
        self.frame_axis = itensor["shape"].index(-1)
        self.factor = self.specified_factor
        if self.axis == self.frame_axis:
            if self.specified_factor is None:
                raise ValueError("Reduce factor must be specified for "
                                 "frame axis")
        else:
            if self.specified_factor is None:
                # default: reduce the whole axis
                self.factor = otensor["shape"][self.axis]
            elif otensor["shape"][self.axis] % self.factor != 0:
                raise ValueError("Reduce factor does not divide axis length")
            otensor["shape"][self.axis] //= self.factor
        otensor["scales"][self.axis][1] *= self.factor
        return ohdr

    def define_output_nframes(self, input_nframe):
        output_nframe = input_nframe
        if self.axis == self.frame_axis:
            if input_nframe % self.factor != 0:
                raise ValueError("Reduce factor does not divide input_nframe")
            output_nframe = input_nframe // self.factor
        return output_nframe

    def on_data(self, ispan, ospan):
        bf_reduce(ispan.data, ospan.data, self.op)


def reduce(iring, axis, factor=None, op="sum", *args, **kwargs):
    """Reduce data along an axis by `factor` using `op`.

    op: sum, mean, min, max, stderr [sum/sqrt(n)], pwrsum [|x|^2 sum],
    pwrmean, pwrmin, pwrmax, pwrstderr.  min/max are not supported for
    complex data.

    Input:  [..., N, ...], dtype = any, space = CUDA
    Output: [..., N/factor, ...], dtype = f32 (cf32 for complex non-pwr)
    """
    return ReduceBlock(iring, axis, factor, op, *args, **kwargs)
