"""CopyBlock: copy between rings/spaces (reference blocks/copy.py surface)."""

from copy import deepcopy

from bifrost_amd.ndarray import copy_array
from bifrost_amd.pipeline import TransformBlock

__all__ = ["CopyBlock", "copy"]


class CopyBlock(TransformBlock):
    def __init__(self, iring, space=None, *args, **kwargs):
        super(CopyBlock, self).__init__(iring, *args, **kwargs)
        if space is None:
            space = self.iring.space
        self.space = space
        self.orings = [self.create_ring(space=space)]

    def define_valid_input_spaces(self):
        return "any"

    def on_sequence(self, iseq):
        return deepcopy(iseq.header)

    def on_data(self, ispan, ospan):
        copy_array(ospan.data, ispan.data)


def copy(iring, space=None, *args, **kwargs):
    """Copy data, possibly between memory spaces.

    Input:  [...], dtype = any, space = any
    Output: [...], dtype = same as input, space = `space`
    """
    return CopyBlock(iring, space, *args, **kwargs)
