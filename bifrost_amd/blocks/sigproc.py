"""SigprocSourceBlock / SigprocSinkBlock (reference blocks/sigproc.py
surface): read and write SIGPROC filterbank / time-series files.

The sink supports the common cases — 3D ['time','pol','freq'] ->
`<name>.fil` filterbank and 2D ['time','pol'] -> `<name>.tim` time
series; the reference's beam- and dispersion- fan-out forms raise
NotImplementedError."""

import os
from copy import copy as _shallow_copy

from bifrost_amd import sigproc
from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import SinkBlock, SourceBlock

__all__ = ["SigprocSourceBlock", "read_sigproc", "SigprocSinkBlock",
           "write_sigproc"]


def _mjd2unix(mjd):
    return (mjd - 40587) * 86400


def _unix2mjd(unix):
    return unix / 86400.0 + 40587


class SigprocSourceBlock(SourceBlock):
    def __init__(self, filenames, gulp_nframe, unpack=True, *args,
                 **kwargs):
        super(SigprocSourceBlock, self).__init__(filenames, gulp_nframe,
                                                 *args, **kwargs)
        self.unpack = unpack

    def create_reader(self, sourcename):
        return sigproc.SigprocFile(sourcename)

    def on_sequence(self, ireader, sourcename):
        ihdr = ireader.header
        assert ihdr["data_type"] in (1, 2, 6)  # filterbank/timeseries/subbands
        for coord_frame in ("pulsarcentric", "barycentric", "topocentric"):
            if coord_frame in ihdr and bool(ihdr[coord_frame]):
                break
        tstart_unix = _mjd2unix(ihdr["tstart"])
        nbit = ihdr["nbits"]
        if self.unpack:
            nbit = max(nbit, 8)
        get = ihdr.get
        ohdr = {
            "_tensor": {
                "dtype": ("i" if ihdr.get("signed") else "u") + str(nbit),
                "shape": [-1, ihdr["nifs"], ihdr["nchans"]],
                "labels": ["time", "pol", "freq"],
                "scales": [(tstart_unix, ihdr["tsamp"]), None,
                           (ihdr["fch1"], ihdr["foff"])],
                "units": ["s", None, "MHz"],
            },
            "frame_rate": 1.0 / ihdr["tsamp"],
            "source_name": get("source_name"),
            "rawdatafile": get("rawdatafile"),
            "az_start": get("az_start"),
            "za_start": get("za_start"),
            "raj": get("src_raj"),
            "dej": get("src_dej"),
            "refdm": get("refdm", 0.0),
            "refdm_units": "pc cm^-3",
            "telescope": sigproc.id2telescope(get("telescope_id")),
            "machine": sigproc.id2machine(get("machine_id")),
            "ibeam": get("ibeam"),
            "nbeams": get("nbeams"),
            "coord_frame": coord_frame,
            "time_tag": int(round(tstart_unix * 2 ** 32)),
            "name": sourcename,
        }
        return [ohdr]

    def on_data(self, reader, ospans):
        ospan = ospans[0]
        if self.unpack:
            indata = reader.read(ospan.data.shape[0])
            nframe = indata.shape[0]
            ospan.data[:nframe] = indata
        else:
            nbyte = reader.readinto(memoryview(ospan.data).cast("B"))
            if nbyte % ospan.frame_nbyte:
                raise IOError("Input file is truncated")
            nframe = nbyte // ospan.frame_nbyte
        return [nframe]


def read_sigproc(filenames, gulp_nframe, unpack=True, *args, **kwargs):
    """Read SIGPROC filterbank/time-series files.

    Output: ['time', 'pol', 'freq'], dtype = u/i*, space = SYSTEM
    """
    return SigprocSourceBlock(filenames, gulp_nframe, unpack, *args,
                              **kwargs)


def _copy_if_exists(dst, src, key, newkey=None):
    if key in src and src[key] is not None:
        dst[newkey or key] = src[key]


class SigprocSinkBlock(SinkBlock):
    def __init__(self, iring, path=None, *args, **kwargs):
        super(SigprocSinkBlock, self).__init__(iring, *args, **kwargs)
        self.path = path or ""
        self.ofile = None

    def on_sequence(self, iseq):
        ihdr = iseq.header
        itensor = ihdr["_tensor"]
        axnames = list(itensor["labels"])
        shape = list(itensor["shape"])
        scales = list(itensor["scales"])
        ndim = len(shape)
        dtype = DataType(itensor["dtype"])

        shdr = {}
        _copy_if_exists(shdr, ihdr, "source_name")
        _copy_if_exists(shdr, ihdr, "rawdatafile")
        _copy_if_exists(shdr, ihdr, "az_start")
        _copy_if_exists(shdr, ihdr, "za_start")
        _copy_if_exists(shdr, ihdr, "raj", "src_raj")
        _copy_if_exists(shdr, ihdr, "dej", "src_dej")
        if ihdr.get("telescope") is not None:
            shdr["telescope_id"] = sigproc.telescope2id(ihdr["telescope"])
        if ihdr.get("machine") is not None:
            shdr["machine_id"] = sigproc.machine2id(ihdr["machine"])
        _copy_if_exists(shdr, ihdr, "ibeam")
        _copy_if_exists(shdr, ihdr, "nbeams")
        shdr["nbits"] = dtype.itemsize_bits
        if dtype.is_integer and dtype.is_signed:
            shdr["signed"] = True
        coord_frame = ihdr.get("coord_frame")
        shdr["pulsarcentric"] = int(coord_frame == "pulsarcentric")
        shdr["barycentric"] = int(coord_frame == "barycentric")

        filename = os.path.join(self.path, ihdr["name"])

        if ndim == 3 and axnames == ["time", "pol", "freq"]:
            assert dtype.is_real
            shdr["data_type"] = 1  # filterbank
            shdr["nifs"] = shape[1]
            shdr["nchans"] = shape[2]
            shdr["tstart"] = _unix2mjd(scales[0][0])
            shdr["tsamp"] = scales[0][1]
            shdr["fch1"] = scales[2][0]
            shdr["foff"] = scales[2][1]
            if ihdr.get("refdm") is not None:
                shdr["refdm"] = ihdr["refdm"]
            filename += ".fil"
        elif ndim == 2 and axnames[0] == "time":
            assert dtype.is_real
            shdr["data_type"] = 2  # time series
            shdr["nchans"] = 1
            shdr["nifs"] = shape[1]
            shdr["tstart"] = _unix2mjd(scales[0][0])
            shdr["tsamp"] = scales[0][1]
            if ihdr.get("refdm") is not None:
                shdr["refdm"] = ihdr["refdm"]
            filename += ".tim"
        else:
            raise NotImplementedError(
                "Unsupported axis layout for sigproc sink: %r" % (axnames,))

        if self.ofile is not None:
            self.ofile.close()
        self.ofile = open(filename, "wb")
        sigproc.write_header(_shallow_copy(shdr), self.ofile)

    def on_sequence_end(self, iseq):
        if self.ofile is not None:
            self.ofile.close()
            self.ofile = None

    def on_data(self, ispan):
        self.ofile.write(ispan.data.tobytes())


def write_sigproc(iring, path=None, *args, **kwargs):
    """Write data as SIGPROC files (.fil filterbank / .tim time series).

    Input: ['time','pol','freq'] or ['time', pol], real dtype, SYSTEM
    """
    return SigprocSinkBlock(iring, path, *args, **kwargs)
