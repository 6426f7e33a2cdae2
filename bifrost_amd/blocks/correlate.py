"""CorrelateBlock: the X step of an FX correlator (reference
blocks/correlate.py:36-136 semantics — cross-multiply stations and
accumulate over nframe_per_integration; lower triangle filled; backed by
the i8-MFMA cherk kernel)."""

from copy import deepcopy

from bifrost_amd.linalg import LinAlg
from bifrost_amd.pipeline import TransformBlock

__all__ = ["CorrelateBlock", "correlate"]


class CorrelateBlock(TransformBlock):
    def __init__(self, iring, nframe_per_integration, *args, **kwargs):
        super(CorrelateBlock, self).__init__(iring, *args, **kwargs)
        self.nframe_per_integration = nframe_per_integration
        self.linalg = LinAlg()

    def define_valid_input_spaces(self):
        return ("cuda",)

    def define_output_nframes(self, input_nframe):
        return 1

    def on_sequence(self, iseq):
        self.nframe_integrated = 0
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        assert itensor["labels"] == ["time", "freq", "station", "pol"]
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]
        otensor["dtype"] = "cf32"
        for key in ("shape", "labels", "scales", "units"):
            if key not in itensor:
                continue
            time_val, freq_val, stand_val, pol_val = itensor[key]
            otensor[key] = [time_val, freq_val, stand_val, pol_val,
                            stand_val, pol_val]
        for i in range(2):
            otensor["labels"][2 + i] += "_i"
            otensor["labels"][4 + i] += "_j"
        if "scales" in otensor and otensor["scales"][0] is not None:
            otensor["scales"][0][1] *= self.nframe_per_integration
        ohdr["matrix_fill_mode"] = "lower"
        ohdr["gulp_nframe"] = min(ohdr["gulp_nframe"],
                                  self.nframe_per_integration)
        gulp_actual = self.gulp_nframe or ohdr["gulp_nframe"]
        if self.nframe_per_integration % gulp_actual != 0:
            raise ValueError(
                "gulp_nframe (%d) does not divide nframe_per_integration (%d)"
                % (gulp_actual, self.nframe_per_integration))
        return ohdr

    def on_data(self, ispan, ospan):
        idata = ispan.data
        odata = ospan.data
        beta = 0 if self.nframe_integrated == 0 else 1
        ntime, nchan, nstand, npol = idata.shape
        idata_mm = idata.reshape([ntime, nchan, nstand * npol]) \
                        .transpose([1, 0, 2])
        odata_mm = odata.reshape([nchan, nstand * npol, nstand * npol])
        assert idata_mm.ctypes.data == idata.ctypes.data
        assert odata_mm.ctypes.data == odata.ctypes.data
        self.linalg.matmul(1, None, idata_mm, beta, odata_mm)
        self.nframe_integrated += ispan.nframe
        assert self.nframe_integrated <= self.nframe_per_integration
        if self.nframe_integrated == self.nframe_per_integration:
            self.nframe_integrated = 0
            return 1
        return 0


def correlate(iring, nframe_per_integration, *args, **kwargs):
    """Cross-multiply stations and accumulate in time (FX correlator X step).

    Input:  ['time','freq','station','pol'], any complex, space = CUDA
    Output: ['time','freq','station_i','pol_i','station_j','pol_j'],
            cf32 (lower triangle filled), space = CUDA
    """
    return CorrelateBlock(iring, nframe_per_integration, *args, **kwargs)
