"""DetectBlock (reference blocks/detect.py surface): square-law detection
of complex voltages, backed by the bfMap JIT."""

import importlib
from copy import deepcopy

from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import TransformBlock

__all__ = ["DetectBlock", "detect"]

_map = importlib.import_module("bifrost_amd.map")


class DetectBlock(TransformBlock):
    def __init__(self, iring, mode, axis=None, *args, **kwargs):
        super(DetectBlock, self).__init__(iring, *args, **kwargs)
        if mode not in ("scalar", "jones", "stokes"):
            raise ValueError("Invalid detect mode: %r" % (mode,))
        if mode != "scalar":
            raise NotImplementedError(
                "only mode='scalar' is implemented this round (DESIGN.md)")
        self.mode = mode
        self.axis = axis

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ohdr = deepcopy(iseq.header)
        itype = DataType(ohdr["_tensor"]["dtype"])
        ohdr["_tensor"]["dtype"] = str(itype.as_real().as_floating_point())
        return ohdr

    def on_data(self, ispan, ospan):
        _map.map("p = c.real*c.real + c.imag*c.imag",
                 {"p": ospan.data, "c": ispan.data})


def detect(iring, mode, axis=None, *args, **kwargs):
    """Square-law detect the input (power of the complex voltages).

    Input:  [...], dtype = any complex, space = CUDA
    Output: [...], dtype = real float, space = CUDA
    """
    return DetectBlock(iring, mode, axis, *args, **kwargs)
