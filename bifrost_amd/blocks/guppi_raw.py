"""GuppiRawSourceBlock (reference blocks/guppi_raw.py surface): stream
GUPPI RAW files into a pipeline as ['time','freq','fine_time','pol'] ci*
blocks."""

from bifrost_amd import guppi_raw
from bifrost_amd.pipeline import SourceBlock

__all__ = ["GuppiRawSourceBlock", "read_guppi_raw"]


def _mjd2unix(mjd):
    return (mjd - 40587) * 86400


class GuppiRawSourceBlock(SourceBlock):
    def __init__(self, sourcenames, gulp_nframe=1, *args, **kwargs):
        super(GuppiRawSourceBlock, self).__init__(
            sourcenames, gulp_nframe=gulp_nframe, *args, **kwargs)

    def create_reader(self, sourcename):
        return open(sourcename, "rb")

    def on_sequence(self, reader, sourcename):
        previous_pos = reader.tell()
        ihdr = guppi_raw.read_header(reader)
        header_size = reader.tell() - previous_pos
        self.header_buf = bytearray(header_size)
        nbit = ihdr["NBITS"]
        assert nbit in (4, 8, 16, 32, 64)
        nchan = ihdr["OBSNCHAN"]
        bw_MHz = ihdr["OBSBW"]
        cfreq_MHz = ihdr["OBSFREQ"]
        df_MHz = bw_MHz / nchan
        f0_MHz = cfreq_MHz - 0.5 * (nchan - 1) * df_MHz
        dt_s = 1.0 / df_MHz / 1e6  # negative when OBSBW < 0: correct
        # timestamp of this block from the packet index
        byte_offset = ihdr["PKTIDX"] * ihdr["PKTSIZE"]
        frame_nbyte = ihdr["BLOCSIZE"] / ihdr["NTIME"]
        bytes_per_sec = frame_nbyte / dt_s
        offset_secs = byte_offset / bytes_per_sec
        tstart_mjd = ihdr["STT_IMJD"] + (ihdr["STT_SMJD"] +
                                         offset_secs) / 86400.0
        tstart_unix = _mjd2unix(tstart_mjd)
        get = ihdr.get
        ohdr = {
            "_tensor": {
                "dtype": "ci" + str(nbit),
                "shape": [-1, nchan, ihdr["NTIME"], ihdr["NPOL"]],
                # 'time' (aka block) is the frame axis
                "labels": ["time", "freq", "fine_time", "pol"],
                "scales": [(tstart_unix, abs(dt_s) * ihdr["NTIME"]),
                           (f0_MHz, df_MHz), (0, dt_s), None],
                "units": ["s", "MHz", "s", None],
                "gulp_nframe": 1,
            },
            "az_start": get("AZ"),
            "za_start": get("ZA"),
            "raj": get("RA") * (24.0 / 360.0) if get("RA") is not None
                   else None,
            "dej": get("DEC"),
            "source_name": get("SRC_NAME"),
            "refdm": get("CHAN_DM"),
            "refdm_units": "pc cm^-3",
            "telescope": get("TELESCOP"),
            "machine": get("BACKEND"),
            "rawdatafile": sourcename,
            "coord_frame": "topocentric",
            # 32 fractional bits (~0.233 ns resolution)
            "time_tag": int(round(tstart_unix * 2 ** 32)),
            "name": sourcename,
        }
        self.already_read_header = True
        return [ohdr]

    def on_data(self, reader, ospans):
        if not self.already_read_header:
            # skip this block's header (same size as the first)
            nbyte = reader.readinto(self.header_buf)
            if nbyte == 0:
                return [0]  # EOF
            if nbyte < len(self.header_buf):
                raise IOError("Block header is truncated")
        self.already_read_header = False
        ospan = ospans[0]
        nbyte = reader.readinto(memoryview(ospan.data).cast("B"))
        if nbyte % ospan.frame_nbyte:
            raise IOError("Block data is truncated")
        return [nbyte // ospan.frame_nbyte]


def read_guppi_raw(filenames, gulp_nframe=1, *args, **kwargs):
    """Read GUPPI RAW files (one block per frame).

    Output: ['time', 'freq', 'fine_time', 'pol'], dtype = ci*, space =
    SYSTEM
    """
    return GuppiRawSourceBlock(filenames, gulp_nframe, *args, **kwargs)
