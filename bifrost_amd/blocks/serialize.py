"""SerializeBlock / DeserializeBlock (reference blocks/serialize.py
surface): stream a sequence to/from a simple on-disk format —
`<name>.bf.json` (the sequence header as JSON) plus `<name>.bf.<frame
offset>.dat` binary data files.

Single-ringlet only (multi-ringlet rings are unsupported here,
DESIGN.md out-of-scope list); the reference's `<name>.bf.<offset>.<ringlet>.dat` form
is recognised but rejected on read.
"""

import glob
import json
import os

from bifrost_amd.pipeline import SinkBlock, SourceBlock

__all__ = ["SerializeBlock", "serialize", "DeserializeBlock", "deserialize"]


def _parse_bf_filename(fname):
    # <basename>.bf.<frame0>[.<ringlet>...].dat
    inds = fname[fname.find(".bf.") + 4:].split(".")[:-1]
    inds = [int(i) for i in inds]
    return inds[0], inds[1:]


class BifrostReader(object):
    """Reader over one serialized sequence (<basename>.bf + .json/.dat)."""

    def __init__(self, basename):
        assert basename.endswith(".bf")
        with open(basename + ".json", "r") as hdr_file:
            self.header = json.load(hdr_file)
        data_filenames = glob.glob(basename + ".*.dat")
        if not data_filenames:
            raise IOError("No data files found for %s" % basename)
        inds = [_parse_bf_filename(f) for f in data_filenames]
        if any(len(ringlets) for _, ringlets in inds):
            raise NotImplementedError("Multi-ringlet serialized data is "
                                      "not supported")
        data_filenames.sort()
        self.files = [open(f, "rb") for f in data_filenames]
        self.cur_file = 0

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        for f in self.files:
            f.close()

    def readinto(self, buf, frame_nbyte):
        """Fill `buf` from the current position, advancing across data
        files; returns whole frames read."""
        view = memoryview(buf).cast("B")
        filled = 0
        while filled < len(view) and self.cur_file < len(self.files):
            nbyte = self.files[self.cur_file].readinto(view[filled:])
            if nbyte == 0:
                self.cur_file += 1
                continue
            filled += nbyte
        if filled % frame_nbyte != 0:
            raise IOError("Unexpected end of file")
        return filled // frame_nbyte


class DeserializeBlock(SourceBlock):
    def __init__(self, filenames, gulp_nframe, *args, **kwargs):
        super(DeserializeBlock, self).__init__(filenames, gulp_nframe,
                                               *args, **kwargs)

    def create_reader(self, sourcename):
        return BifrostReader(sourcename)

    def on_sequence(self, ireader, sourcename):
        return [ireader.header]

    def on_data(self, reader, ospans):
        ospan = ospans[0]
        return [reader.readinto(ospan.data, ospan.frame_nbyte)]


def deserialize(filenames, gulp_nframe, *args, **kwargs):
    """Deserialize a data stream from files written by `serialize`.

    Input:  one `<basename>.bf` per sequence (.json header + .dat data)
    Output: [frame, ...], dtype = any, space = SYSTEM
    """
    return DeserializeBlock(filenames, gulp_nframe, *args, **kwargs)


class SerializeBlock(SinkBlock):
    def __init__(self, iring, path=None, max_file_size=None, *args,
                 **kwargs):
        super(SerializeBlock, self).__init__(iring, *args, **kwargs)
        self.path = path or ""
        if max_file_size is None:
            max_file_size = 1024 ** 3
        self.max_file_size = max_file_size
        self.ofile = None

    def _close_data_file(self):
        if self.ofile is not None:
            self.ofile.close()
            self.ofile = None

    def _open_new_data_file(self, frame_offset):
        self._close_data_file()
        self.bytes_written = 0
        self.ofile = open(self.basename + ".bf.%012i.dat" % frame_offset,
                          "wb")

    def on_sequence(self, iseq):
        hdr = iseq.header
        tensor = hdr["_tensor"]
        if hdr.get("name"):
            self.basename = hdr["name"]
        else:
            self.basename = "%020i" % hdr["time_tag"]
        if self.path:
            self.basename = os.path.join(self.path,
                                         os.path.basename(self.basename))
        with open(self.basename + ".bf.json", "w") as hdr_file:
            hdr_file.write(json.dumps(hdr, indent=4, sort_keys=True))
        if tensor["shape"].index(-1) != 0:
            raise NotImplementedError("Ringlet serialization is not "
                                      "supported")
        self._open_new_data_file(frame_offset=0)

    def on_sequence_end(self, iseq):
        self._close_data_file()

    def on_data(self, ispan):
        data = ispan.data
        nbyte = data.nbytes
        if self.max_file_size > 0 and \
                self.bytes_written + nbyte > self.max_file_size and \
                self.bytes_written > 0:
            self._open_new_data_file(ispan.frame_offset)
        self.bytes_written += nbyte
        data.tofile(self.ofile)


def serialize(iring, path=None, max_file_size=None, *args, **kwargs):
    """Serialize a data stream to `<name>.bf.json` + `<name>.bf.*.dat`.

    Filenames begin with the sequence name, or the time tag if unnamed.
    `max_file_size` rolls data files at that many bytes (-1 = no limit).

    Input:  [frame, ...], dtype = any, space = SYSTEM
    Output: one data file (set) per sequence
    """
    return SerializeBlock(iring, path, max_file_size, *args, **kwargs)
