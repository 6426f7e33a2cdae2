"""BinaryFileReadBlock / BinaryFileWriteBlock (reference
blocks/binary_io.py surface): raw binary file streaming for pipeline
testing and replay."""

import numpy as np

from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import SinkBlock, SourceBlock

__all__ = ["BinaryFileReadBlock", "BinaryFileWriteBlock", "binary_read",
           "binary_write"]


class BinaryFileRead(object):
    """File-like reader yielding gulp_size-element numpy chunks."""

    def __init__(self, filename, gulp_size, dtype):
        self.file_obj = open(filename, "rb")
        self.dtype = dtype
        self.gulp_size = gulp_size

    def read(self):
        return np.fromfile(self.file_obj, dtype=self.dtype,
                           count=self.gulp_size)

    def close(self):
        self.file_obj.close()

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.close()


class BinaryFileReadBlock(SourceBlock):
    def __init__(self, filenames, gulp_size, gulp_nframe, dtype, *args,
                 **kwargs):
        super(BinaryFileReadBlock, self).__init__(filenames, gulp_nframe,
                                                  *args, **kwargs)
        self.dtype = dtype
        self.gulp_size = gulp_size

    def create_reader(self, filename):
        np_dtype = DataType(self.dtype).as_numpy_dtype()
        return BinaryFileRead(filename, self.gulp_size, np_dtype)

    def on_sequence(self, ireader, filename):
        ohdr = {
            "name": filename,
            "_tensor": {
                "dtype": self.dtype,
                "shape": [-1, self.gulp_size],
                "labels": ["streamed", "gulped"],
                "units": [None, None],
                "scales": [[0, 1], [0, 1]],
            },
        }
        return [ohdr]

    def on_data(self, reader, ospans):
        indata = reader.read()
        if indata.shape[0] == self.gulp_size:
            ospans[0].data[0] = indata
            return [1]
        return [0]


class BinaryFileWriteBlock(SinkBlock):
    def __init__(self, iring, file_ext="out", *args, **kwargs):
        super(BinaryFileWriteBlock, self).__init__(iring, *args, **kwargs)
        self.current_fileobj = None
        self.file_ext = file_ext

    def __del__(self):
        try:
            self.current_fileobj.close()
        except AttributeError:
            pass

    def on_sequence(self, iseq):
        if self.current_fileobj is not None:
            self.current_fileobj.close()
        self.current_fileobj = open(iseq.header["name"] + "." +
                                    self.file_ext, "wb")

    def on_sequence_end(self, iseq):
        if self.current_fileobj is not None:
            self.current_fileobj.close()
            self.current_fileobj = None

    def on_data(self, ispan):
        self.current_fileobj.write(ispan.data.tobytes())


def binary_read(filenames, gulp_size, gulp_nframe, dtype, *args, **kwargs):
    """Stream raw binary files into a pipeline.

    Output: ['streamed', 'gulped'] of shape [-1, gulp_size], system space.
    """
    return BinaryFileReadBlock(filenames, gulp_size, gulp_nframe, dtype,
                               *args, **kwargs)


def binary_write(iring, file_ext="out", *args, **kwargs):
    """Write ring data to `<sequence name>.<file_ext>` binary files."""
    return BinaryFileWriteBlock(iring, file_ext, *args, **kwargs)
