"""UnpackBlock (reference blocks/unpack.py surface): unpack sub-byte
samples to ci8/i8."""

from copy import deepcopy

import bifrost_amd as bf
from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import TransformBlock

__all__ = ["UnpackBlock", "unpack"]


class UnpackBlock(TransformBlock):
    def __init__(self, iring, dtype, align_msb=False, *args, **kwargs):
        super(UnpackBlock, self).__init__(iring, *args, **kwargs)
        self.dtype = dtype
        self.align_msb = align_msb

    def define_valid_input_spaces(self):
        return "any"

    def on_sequence(self, iseq):
        ohdr = deepcopy(iseq.header)
        itype = DataType(ohdr["_tensor"]["dtype"])
        self.itype = itype
        otype = DataType(self.dtype)
        ohdr["_tensor"]["dtype"] = str(otype)
        return ohdr

    def on_data(self, ispan, ospan):
        bf.unpack(ispan.data, ospan.data, self.align_msb)


def unpack(iring, dtype, *args, **kwargs):
    """Unpack 1, 2 or 4-bit data to 8-bit.

    Input:  [...], dtype = ci4, space = any
    Output: [...], dtype = ci8 (or i8), space = same as input
    """
    return UnpackBlock(iring, dtype, *args, **kwargs)
