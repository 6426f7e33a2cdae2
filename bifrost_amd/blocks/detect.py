"""DetectBlock (reference blocks/detect.py surface): square-law detection
of complex voltages into polarization products, backed by the bfMap JIT.

Modes (reference blocks/detect.py:141-149):
  scalar:   x   -> real |x|^2
  jones:    x,y -> complex |x|^2 + 1j|y|^2, x.y*
  stokes:   x,y -> real I, Q, U, V
  stokes_i: x,y -> real I = |x|^2 + |y|^2
  coherence:x,y -> real |x|^2, |y|^2, Re(x*y), Im(x*y)
"""

import importlib
from copy import deepcopy

from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import TransformBlock

__all__ = ["DetectBlock", "detect"]

_map = importlib.import_module("bifrost_amd.map")


class DetectBlock(TransformBlock):
    def __init__(self, iring, mode, axis=None, *args, **kwargs):
        super(DetectBlock, self).__init__(iring, *args, **kwargs)
        self.specified_axis = axis
        self.mode = mode.lower()
        if self.mode not in ("scalar", "jones", "stokes", "stokes_i",
                             "coherence"):
            raise ValueError("Invalid detect mode: %r" % (mode,))

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        itype = DataType(itensor["dtype"])
        if not itype.is_complex:
            raise TypeError("Input data must be complex")
        self.axis = self.specified_axis
        if "labels" not in itensor and self.axis is None:
            raise TypeError("Polarization (pol) axis must be labelled, or "
                            "axis must be set manually")
        elif (self.axis is None and self.mode != "scalar" and
              "pol" in itensor["labels"]):
            self.axis = itensor["labels"].index("pol")
        elif isinstance(self.axis, str):
            self.axis = itensor["labels"].index(self.axis)
        # axis None => single-pol mode
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]
        if self.axis is not None:
            self.npol = otensor["shape"][self.axis]
            if self.npol not in (1, 2):
                raise ValueError("Axis must have length 1 or 2")
            if self.mode in ("stokes", "coherence") and self.npol == 2:
                otensor["shape"][self.axis] = 4
            if self.mode == "stokes_i" and self.npol == 2:
                otensor["shape"][self.axis] = 1
            if "labels" in otensor:
                otensor["labels"][self.axis] = "pol"
        else:
            self.npol = 1
        if self.mode == "jones" and self.npol == 2:
            otype = itype
        else:
            otype = itype.as_real()
        otensor["dtype"] = str(otype.as_floating_point())
        return ohdr

    def on_data(self, ispan, ospan):
        idata = ispan.data
        odata = ospan.data
        if self.npol == 1:
            _map.map("b = Complex<b_type>(a).mag2()",
                     {"a": idata, "b": odata})
            return
        shape = idata.shape[:self.axis] + idata.shape[self.axis + 1:]
        inds = ["i%i" % i for i in range(idata.ndim)]
        inds[self.axis] = "%i"
        inds_pol = ",".join(inds)
        inds_ = [inds_pol % i for i in range(4)]
        inds = inds[:self.axis] + inds[self.axis + 1:]
        if self.mode == "jones":
            func = """
            b_type x = a(%s);
            b_type y = a(%s);
            b(%s).assign(x.mag2(), y.mag2());
            b(%s) = x*y.conj();
            """ % (inds_[0], inds_[1], inds_[0], inds_[1])
        elif self.mode == "stokes":
            func = """
            Complex<b_type> x = a(%s);
            Complex<b_type> y = a(%s);
            auto xx = x.mag2();
            auto yy = y.mag2();
            auto xy = x*y.conj();
            b(%s) = xx + yy;
            b(%s) = xx - yy;
            b(%s) =  2*xy.real;
            b(%s) = -2*xy.imag;
            """ % (inds_[0], inds_[1],
                   inds_[0], inds_[1], inds_[2], inds_[3])
        elif self.mode == "stokes_i":
            func = """
            Complex<b_type> x = a(%s);
            Complex<b_type> y = a(%s);
            b(%s) = x.mag2() + y.mag2();
            """ % (inds_[0], inds_[1], inds_[0])
        elif self.mode == "coherence":
            func = """
            Complex<b_type> x = a(%s);
            Complex<b_type> y = a(%s);
            auto xy = x.conj()*y;
            b(%s) = x.mag2();
            b(%s) = y.mag2();
            b(%s) = xy.real;
            b(%s) = xy.imag;
            """ % (inds_[0], inds_[1],
                   inds_[0], inds_[1], inds_[2], inds_[3])
        else:
            raise ValueError(self.mode)
        _map.map(func, shape=shape, axis_names=inds,
                 data={"a": idata, "b": odata})


def detect(iring, mode, axis=None, *args, **kwargs):
    """Apply square-law detection to create polarization products.

    mode: 'scalar' | 'jones' | 'stokes' | 'stokes_i' | 'coherence';
    axis names the polarization axis (defaults to the 'pol' label).

    Input:  [..., 'pol', ...], dtype = any complex, space = CUDA
    Output: [..., 'pol', ...], dtype = real or complex float, space = CUDA
    """
    return DetectBlock(iring, mode, axis, *args, **kwargs)
