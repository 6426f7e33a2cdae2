"""DadaFileReadBlock (reference blocks/dada_file.py surface): stream
PSRDADA disk files — a 4096-byte ASCII "KEY  value" header followed by
raw binary data — into a pipeline.  The caller supplies header_callback
to build the bifrost '_tensor' header from the free-form DADA keywords.
"""

import glob
import os

import numpy as np

from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import SourceBlock

__all__ = ["DadaFileRead", "DadaFileReadBlock", "read_dada_file",
           "generate_dada_filelist"]

_HEADER_BYTES = 4096


class DadaFileRead(object):
    """Reader over one or more DADA files (frames along axis 0)."""

    def __init__(self, filename, header_callback):
        if isinstance(filename, str):
            self.filenames = [filename]
        else:
            self.filenames = list(filename)
        self.nfiles = len(self.filenames)
        self.fcount = 0
        self.file_obj = open(self.filenames[0], "rb")
        self._header_callback = header_callback
        self.header = self._read_header()
        itensor = self.header["_tensor"]
        self.dtype = DataType(itensor["dtype"]).as_numpy_dtype()
        self.block_shape = list(itensor["shape"])
        self.block_shape[0] = 1
        self.block_size = int(np.prod(self.block_shape))

    def _read_header(self):
        raw = self.file_obj.read(_HEADER_BYTES).decode("ascii",
                                                       errors="replace")
        hdr = {}
        for line in raw.split("\n"):
            parts = line.split()
            if len(parts) == 2:
                hdr[parts[0]] = parts[1]
        return self._header_callback(hdr)

    def _open_next_file(self):
        self.file_obj.close()
        self.fcount += 1
        self.file_obj = open(self.filenames[self.fcount], "rb")
        self._read_header()  # skip (assumed identical)

    def read(self):
        """Read one frame; returns an empty array at EOF."""
        d = np.fromfile(self.file_obj, dtype=self.dtype,
                        count=self.block_size)
        if d.size == 0 and self.fcount < self.nfiles - 1:
            self._open_next_file()
            d = np.fromfile(self.file_obj, dtype=self.dtype,
                            count=self.block_size)
        if d.size == 0:
            return d
        if d.size != self.block_size:
            raise IOError("DADA file truncated mid-frame")
        return d.reshape(self.block_shape)

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.file_obj.close()


def generate_dada_filelist(filename):
    """List all DADA files belonging to the capture `filename` starts
    (strips the trailing _<offset>.<...>.dada part and globs)."""
    bn = os.path.basename(filename)
    dn = os.path.dirname(filename)
    bn_root = "_".join(bn.split("_")[:-1])
    return sorted(glob.glob(os.path.join(dn, bn_root + "_*.dada")))


class DadaFileReadBlock(SourceBlock):
    def __init__(self, filename, header_callback, gulp_nframe, *args,
                 **kwargs):
        super(DadaFileReadBlock, self).__init__(filename, gulp_nframe,
                                                *args, **kwargs)
        self.header_callback = header_callback

    def create_reader(self, filename):
        flist = generate_dada_filelist(filename)
        return DadaFileRead(flist, self.header_callback)

    def on_sequence(self, ireader, filename):
        return [ireader.header]

    def on_data(self, reader, ospans):
        indata = reader.read()
        if indata.size == 0:
            return [0]
        ospans[0].data[0] = indata[0]
        return [1]


def read_dada_file(filename, header_callback, gulp_nframe, *args,
                   **kwargs):
    """Read a list of DADA files into a pipeline.

    header_callback(dada_hdr_dict) must return a bifrost sequence header
    with a '_tensor' (shape[0] = -1 frame axis).
    """
    return DadaFileReadBlock(filename, header_callback, gulp_nframe,
                             *args, **kwargs)
