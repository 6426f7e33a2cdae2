"""AccumulateBlock (reference blocks/accumulate.py surface): sum frames
over an integration window."""

from copy import deepcopy

import numpy as np

from bifrost_amd.pipeline import TransformBlock

__all__ = ["AccumulateBlock", "accumulate"]


class AccumulateBlock(TransformBlock):
    def __init__(self, iring, nframe_to_accumulate, *args, **kwargs):
        super(AccumulateBlock, self).__init__(iring, *args, **kwargs)
        self.nframe_to_accumulate = nframe_to_accumulate
        kwargs.setdefault("gulp_nframe", 1)

    def define_valid_input_spaces(self):
        return ("system",)

    def define_output_nframes(self, input_nframe):
        return 1

    def on_sequence(self, iseq):
        self.nframe_accumulated = 0
        ohdr = deepcopy(iseq.header)
        if "scales" in ohdr["_tensor"] and ohdr["_tensor"]["scales"][0]:
            ohdr["_tensor"]["scales"][0][1] *= self.nframe_to_accumulate
        return ohdr

    def on_data(self, ispan, ospan):
        idata = np.asarray(ispan.data)
        odata = np.asarray(ospan.data)
        summed = idata.sum(axis=0, keepdims=True)
        if self.nframe_accumulated == 0:
            odata[...] = summed
        else:
            odata[...] += summed
        self.nframe_accumulated += ispan.nframe
        assert self.nframe_accumulated <= self.nframe_to_accumulate
        if self.nframe_accumulated == self.nframe_to_accumulate:
            self.nframe_accumulated = 0
            return 1
        return 0


def accumulate(iring, nframe_to_accumulate, *args, **kwargs):
    """Accumulate (sum) frames over a window.

    Input:  [...], any numeric dtype, space = system
    Output: [...], same dtype, 1 frame per window
    """
    return AccumulateBlock(iring, nframe_to_accumulate, *args, **kwargs)
