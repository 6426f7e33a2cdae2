"""ScrunchBlock (reference blocks/scrunch.py surface): average `factor`
frames into one, on system memory.  (Deprecated upstream in favour of
ReduceBlock; kept for API parity.)"""

from copy import deepcopy

from bifrost_amd.pipeline import TransformBlock

__all__ = ["ScrunchBlock", "scrunch"]


class ScrunchBlock(TransformBlock):
    def __init__(self, iring, factor, *args, **kwargs):
        super(ScrunchBlock, self).__init__(iring, *args, **kwargs)
        assert isinstance(factor, int)
        self.factor = factor

    def define_valid_input_spaces(self):
        return ("system",)

    def define_output_nframes(self, input_nframe):
        if input_nframe % self.factor != 0:
            raise ValueError("Scrunch factor does not divide gulp size")
        return input_nframe // self.factor

    def on_sequence(self, iseq):
        ohdr = deepcopy(iseq.header)
        ohdr["_tensor"]["scales"][0][1] *= self.factor
        return ohdr

    def on_data(self, ispan, ospan):
        import numpy as np
        in_nframe = ispan.nframe
        out_nframe = in_nframe // self.factor
        idata = np.asarray(ispan.data)
        odata = np.asarray(ospan.data)
        odata[...] = idata.reshape((out_nframe, self.factor) +
                                   idata.shape[1:]) \
                          .mean(axis=1, dtype=odata.dtype)
        return out_nframe


def scrunch(iring, factor, *args, **kwargs):
    """Average `factor` incoming frames into one output frame (SYSTEM
    space)."""
    return ScrunchBlock(iring, factor, *args, **kwargs)
