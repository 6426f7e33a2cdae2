"""FftBlock (reference blocks/fft.py surface): FFT along named axes."""

from copy import deepcopy

from bifrost_amd.DataType import DataType
from bifrost_amd.fft import Fft
from bifrost_amd.pipeline import TransformBlock

__all__ = ["FftBlock", "fft"]


class FftBlock(TransformBlock):
    def __init__(self, iring, axes, inverse=False, real_output=False,
                 axis_labels=None, *args, **kwargs):
        super(FftBlock, self).__init__(iring, *args, **kwargs)
        if not isinstance(axes, (list, tuple)):
            axes = [axes]
        self.specified_axes = axes
        self.inverse = inverse
        self.real_output = real_output
        self.axis_labels = axis_labels
        self.fft = Fft()
        self._plan_key = None

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        axes = []
        for a in self.specified_axes:
            if not isinstance(a, int):
                a = itensor["labels"].index(a)
            axes.append(a)
        self.axes = axes
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]
        itype = DataType(itensor["dtype"])
        if self.real_output:
            otensor["dtype"] = str(itype.as_real())
            # c2r: real length = 2*(n-1) on the last transformed axis
            last = axes[-1]
            otensor["shape"][last] = 2 * (itensor["shape"][last] - 1)
        elif itype.is_real:
            otensor["dtype"] = str(itype.as_complex())
            last = axes[-1]
            otensor["shape"][last] = itensor["shape"][last] // 2 + 1
        if self.axis_labels is not None:
            for a, lbl in zip(axes, self.axis_labels):
                otensor["labels"][a] = lbl
        self._plan_key = None
        return ohdr

    def on_data(self, ispan, ospan):
        idata = ispan.data
        odata = ospan.data
        key = tuple(idata.shape)
        if key != self._plan_key:
            self.fft.init(idata, odata, axes=self.axes)
            self._plan_key = key
        self.fft.execute(idata, odata, inverse=self.inverse)


def fft(iring, axes, inverse=False, real_output=False, axis_labels=None,
        *args, **kwargs):
    """Apply an FFT along the given axes.

    Input:  [...], dtype = [c]f32/[c]f64, space = CUDA
    Output: [...], transformed along `axes` (unnormalized)
    """
    return FftBlock(iring, axes, inverse, real_output, axis_labels,
                    *args, **kwargs)
