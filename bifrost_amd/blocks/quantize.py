"""QuantizeBlock (reference blocks/quantize.py surface): quantize floats
to (complex) integers."""

from copy import deepcopy

import bifrost_amd as bf
from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import TransformBlock

__all__ = ["QuantizeBlock", "quantize"]


class QuantizeBlock(TransformBlock):
    def __init__(self, iring, dtype, scale=1.0, *args, **kwargs):
        super(QuantizeBlock, self).__init__(iring, *args, **kwargs)
        self.dtype = dtype
        self.scale = scale

    def define_valid_input_spaces(self):
        return "any"

    def on_sequence(self, iseq):
        ohdr = deepcopy(iseq.header)
        otype = DataType(self.dtype)
        ohdr["_tensor"]["dtype"] = str(otype)
        return ohdr

    def on_data(self, ispan, ospan):
        bf.quantize(ispan.data, ospan.data, self.scale)


def quantize(iring, dtype, scale=1.0, *args, **kwargs):
    """Quantize data to the specified integer dtype.

    Input:  [...], dtype = [c]f32, space = any
    Output: [...], dtype = any (complex) integer type, space = same
    """
    return QuantizeBlock(iring, dtype, scale, *args, **kwargs)
