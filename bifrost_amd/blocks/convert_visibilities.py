"""ConvertVisibilitiesBlock (reference blocks/convert_visibilities.py
surface): convert correlator visibilities between formats —

  'matrix'  : ['time','freq','station_i','pol_i','station_j','pol_j'],
              Hermitian; produced lower-filled by CorrelateBlock
              (matrix_fill_mode='lower'), fmt='matrix' fills the upper
              triangle by conjugation.
  'storage' : ['time','baseline','freq','stokes'], the lower-triangle
              baseline list with Stokes products (UVFITS/MS-style
              ordering, baseline b = i(i+1)/2 + j).

Runs on the bfMap indexed form with 2-/4-vector complex dtypes.
"""

import importlib
from copy import deepcopy
from math import sqrt

from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import TransformBlock

__all__ = ["ConvertVisibilitiesBlock", "convert_visibilities"]

_map = importlib.import_module("bifrost_amd.map")

_MATRIX_LABELS = ["freq", "station_i", "pol_i", "station_j", "pol_j"]

# matrix(lower) -> matrix(full): mirror the strict upper triangle from
# the conjugate transpose; on the diagonal, pol block [i,1,i,0] mirrors
# [i,0,i,1]* (only the lower pol product is filled there).
_FILL_FUNC = """
if (i > j) {
    odata(t,c,i,0,j,0) = idata(t,c,i,0,j,0);
    odata(t,c,i,1,j,0) = idata(t,c,i,1,j,0);
} else {
    auto x = idata(t,c,j,0,i,0);
    auto y = idata(t,c,j,1,i,0);
    auto xy = x[1];
    x[0] = x[0].conj();
    x[1] = y[0].conj();
    if (i != j) y[0] = xy.conj();
    y[1] = y[1].conj();
    odata(t,c,i,0,j,0) = x;
    odata(t,c,i,1,j,0) = y;
}
"""

# matrix(lower) -> storage: unrank baseline b -> (i,j), then Stokes.
# (float sqrt unranking is exact up to ~2048 stations, as the reference
# notes.)
_TO_STORAGE_FUNC = """
int i = int((sqrt(8.f*(b)+1)-1)/2);
int j = b - i*(i+1)/2;
auto x = idata(t,c,i,0,j,0);
auto y = idata(t,c,i,1,j,0);
if (i == j) x[1] = y[0].conj();
idata_type::value_type eye(0, 1);
auto I = x[0] + y[1];
auto Q = x[0] - y[1];
auto U = x[1] + y[0];
auto V = (x[1] - y[0]) * eye;
odata(t,b,c,0) = odata_type(I, Q, U, V);
"""

# storage -> matrix(full): rank (i,j) -> b (swapping to the lower pair),
# reconstruct the pol block from IQUV, conjugating for the upper output.
_TO_MATRIX_FUNC = """
bool upper = (i < j);
auto b = upper ? j*(j+1)/2 + i : i*(i+1)/2 + j;
auto S = idata(t,b,c,0);
idata_type::value_type eye(0, 1);
auto xx = 0.5f*(S[0] + S[1]);
auto xy = 0.5f*(S[2] - S[3]*eye);
auto yx = 0.5f*(S[2] + S[3]*eye);
auto yy = 0.5f*(S[0] - S[1]);
if (i == j) xy = yx.conj();
if (upper) {
    auto t2 = xy;
    xx = xx.conj();
    xy = yx.conj();
    yx = t2.conj();
    yy = yy.conj();
}
odata(t,c,i,0,j,0) = odata_type(xx, xy);
odata(t,c,i,1,j,0) = odata_type(yx, yy);
"""


class ConvertVisibilitiesBlock(TransformBlock):
    def __init__(self, iring, fmt, *args, **kwargs):
        super(ConvertVisibilitiesBlock, self).__init__(iring, *args,
                                                       **kwargs)
        self.ofmt = fmt

    def define_valid_input_spaces(self):
        return ("cuda",)

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        ilabels = itensor["labels"]
        assert ilabels[0] == "time"
        ohdr = deepcopy(ihdr)
        otensor = ohdr["_tensor"]

        if ilabels[1:] == _MATRIX_LABELS:
            nchan, nstand, npol, nstand_j, npol_j = itensor["shape"][1:]
            assert nstand_j == nstand
            assert npol_j == npol
            self.ifmt = "matrix"
            if self.ofmt == "matrix":
                ohdr["matrix_fill_mode"] = "hermitian"
            elif self.ofmt == "storage":
                nbaseline = nstand * (nstand + 1) // 2
                ohdr.pop("matrix_fill_mode", None)
                otensor["labels"] = ["time", "baseline", "freq", "stokes"]
                otensor["shape"] = [-1, nbaseline, nchan, npol * npol]
                units = itensor.get("units")
                if units:
                    otensor["units"] = [units[0], None, units[1],
                                        ("I", "Q", "U", "V")]
            else:
                raise NotImplementedError(
                    "Unsupported conversion from %s to %s"
                    % (self.ifmt, self.ofmt))
        elif ilabels[1:] == ["baseline", "freq", "stokes"]:
            nbaseline, nchan, nstokes = itensor["shape"][1:]
            assert nstokes in (1, 4)
            npol = 1 if nstokes == 1 else 2
            nstand = int(sqrt(8 * nbaseline + 1) - 1) // 2
            self.ifmt = "storage"
            if self.ofmt == "matrix":
                otensor["labels"] = ["time"] + _MATRIX_LABELS
                otensor["shape"] = [-1, nchan, nstand, npol, nstand, npol]
                units = itensor.get("units")
                if units:
                    pol_units = ("X", "Y")
                    otensor["units"] = [units[0], units[2], None,
                                        pol_units, None, pol_units]
            else:
                raise NotImplementedError(
                    "Unsupported conversion from %s to %s"
                    % (self.ifmt, self.ofmt))
        else:
            raise NotImplementedError("Cannot convert input from %s to %s"
                                      % (ilabels, self.ofmt))
        return ohdr

    def on_data(self, ispan, ospan):
        idata = ispan.data
        odata = ospan.data
        itype = DataType(idata.dtype)
        otype = DataType(odata.dtype)
        if self.ifmt == "matrix" and self.ofmt == "matrix":
            shape = list(idata.shape)
            del shape[5]
            del shape[3]
            idata = idata.view(itype.as_vector(2))
            odata = odata.view(otype.as_vector(2))
            _map.map(_FILL_FUNC, shape=shape,
                     axis_names=["t", "c", "i", "j"],
                     data={"idata": idata, "odata": odata})
        elif self.ifmt == "matrix" and self.ofmt == "storage":
            assert idata.shape[2] <= 2048  # float sqrt unranking limit
            idata = idata.view(itype.as_vector(2))
            odata = odata.view(otype.as_vector(4))
            _map.map(_TO_STORAGE_FUNC, shape=odata.shape[:-1],
                     axis_names=["t", "b", "c"],
                     data={"idata": idata, "odata": odata})
        elif self.ifmt == "storage" and self.ofmt == "matrix":
            shape = list(odata.shape)
            del shape[5]
            del shape[3]
            idata = idata.view(itype.as_vector(4))
            odata = odata.view(otype.as_vector(2))
            _map.map(_TO_MATRIX_FUNC, shape=shape,
                     axis_names=["t", "c", "i", "j"],
                     data={"idata": idata, "odata": odata})
        else:
            raise NotImplementedError


def convert_visibilities(iring, fmt, *args, **kwargs):
    """Convert visibility data between 'matrix' and 'storage' formats.

    Input:  ['time','freq','station_i','pol_i','station_j','pol_j'] or
            ['time','baseline','freq','stokes'], complex, space = CUDA
    Output: per `fmt` (see class docstring)
    """
    return ConvertVisibilitiesBlock(iring, fmt, *args, **kwargs)
