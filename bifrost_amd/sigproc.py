"""SIGPROC filterbank/time-series file format (reference
python/bifrost/sigproc.py + sigproc2.py surfaces; the on-disk format is
the SIGPROC standard: '=i'-length-prefixed keys between HEADER_START and
HEADER_END, with int ('=i'), double ('=d'), char ('=b') or
length-prefixed string values; data follows as [time][if/pol][chan]).
"""

import os
import struct
import warnings

import numpy as np

__all__ = ["read_header", "write_header", "seek_to_data", "pack", "unpack",
           "id2telescope", "telescope2id", "id2machine", "machine2id",
           "SigprocFile"]

_STRING_VALUES = ["source_name", "rawdatafile"]
_DOUBLE_VALUES = ["az_start", "za_start", "src_raj", "src_dej", "tstart",
                  "tsamp", "period", "fch1", "foff", "refdm"]
_INTEGER_VALUES = ["nchans", "telescope_id", "machine_id", "data_type",
                   "ibeam", "nbeams", "nbits", "barycentric",
                   "pulsarcentric", "nbins", "nsamples", "nifs", "npuls"]
_CHARACTER_VALUES = ["signed"]

_TELESCOPES = {0: "Fake", 1: "Arecibo", 2: "Ooty", 3: "Nancay", 4: "Parkes",
               5: "Jodrell", 6: "GBT", 7: "GMRT", 8: "Effelsberg",
               9: "Effelsberg LOFAR", 11: "Unknown", 12: "MWA", 20: "CHIME",
               52: "LWA-OV", 53: "LWA-SV", 64: "MeerKAT", 65: "KAT-7",
               82: "eMerlin"}
_MACHINES = {0: "FAKE", 1: "PSPM", 2: "WAPP", 3: "AOFTM", 4: "BPP",
             5: "OOTY", 6: "SCAMP", 7: "GMRTFB", 8: "PULSAR2000",
             9: "UNKNOWN", 20: "CHIME", 52: "LWA-DP", 53: "LWA-ADP"}
_TELESCOPE_IDS = {v: k for k, v in _TELESCOPES.items()}
_MACHINE_IDS = {v: k for k, v in _MACHINES.items()}


def id2telescope(tid):
    return _TELESCOPES.get(tid, "unknown")


def telescope2id(name):
    return _TELESCOPE_IDS[name]


def id2machine(mid):
    return _MACHINES.get(mid, "unknown")


def machine2id(name):
    return _MACHINE_IDS[name]


def _write_key(f, key):
    f.write(struct.pack("=i", len(key)))
    f.write(key.encode())


def write_header(hdr, f):
    """Write a sigproc header dict at the current file position."""
    _write_key(f, "HEADER_START")
    for key, val in hdr.items():
        if key in _STRING_VALUES:
            _write_key(f, key)
            _write_key(f, val)
        elif key in _DOUBLE_VALUES:
            _write_key(f, key)
            f.write(struct.pack("=d", float(val)))
        elif key in _INTEGER_VALUES:
            _write_key(f, key)
            f.write(struct.pack("=i", int(val)))
        elif key in _CHARACTER_VALUES:
            _write_key(f, key)
            f.write(struct.pack("=b", int(val)))
        elif key == "header_size":
            pass
        else:
            warnings.warn("Unknown sigproc header key: %r" % (key,),
                          RuntimeWarning)
    _write_key(f, "HEADER_END")


def _read_key(f):
    raw = f.read(4)
    if len(raw) < 4:
        return None
    length = struct.unpack("=i", raw)[0]
    if length <= 0 or length >= 80:
        return None
    return f.read(length).decode()


def read_header(f):
    """Read a sigproc header from the start of `f`; returns a dict with
    'header_size' set to the data offset."""
    f.seek(0)
    if _read_key(f) != "HEADER_START":
        f.seek(0)
        raise ValueError("Missing HEADER_START")
    header = {}
    expecting = None
    while True:
        key = _read_key(f)
        if key is None:
            raise ValueError("Failed to parse header")
        if key == "HEADER_END":
            break
        elif key in _STRING_VALUES:
            expecting = key
        elif key in _DOUBLE_VALUES:
            header[key] = struct.unpack("=d", f.read(8))[0]
        elif key in _INTEGER_VALUES:
            header[key] = struct.unpack("=i", f.read(4))[0]
        elif key in _CHARACTER_VALUES:
            header[key] = struct.unpack("=b", f.read(1))[0]
        elif expecting is not None:
            header[expecting] = key
            expecting = None
        else:
            warnings.warn("Unknown header key: %r" % (key,), RuntimeWarning)
    if "nchans" not in header:
        header["nchans"] = 1
    header["header_size"] = f.tell()
    return header


def seek_to_data(f):
    """Position `f` at the first data byte."""
    read_header(f)


def pack(data, nbit):
    """Pack 8-bit values down to nbit (LSB-first within each byte)."""
    data = np.asarray(data).flatten()
    if 8 % nbit != 0:
        raise ValueError("pack: nbit must divide into 8")
    if data.dtype not in (np.uint8, np.int8):
        raise TypeError("pack: dtype must be 8-bit")
    vals_per_byte = 8 // nbit
    mask = (1 << nbit) - 1
    out = np.zeros(data.size // vals_per_byte, dtype=np.uint8)
    for i in range(vals_per_byte):
        out |= (data[i::vals_per_byte].astype(np.uint8) & mask) << (nbit * i)
    return out


def unpack(data, nbit):
    """Unpack nbit-packed bytes up to 8-bit values (LSB-first)."""
    data = np.asarray(data)
    if nbit > 8:
        raise ValueError("unpack: nbit must be <= 8")
    if 8 % nbit != 0:
        raise ValueError("unpack: nbit must divide into 8")
    if data.dtype not in (np.uint8, np.int8):
        raise TypeError("unpack: dtype must be 8-bit")
    if nbit == 8:
        return data
    vals_per_byte = 8 // nbit
    mask = (1 << nbit) - 1
    raw = data.view(np.uint8).flatten()
    out = np.empty(raw.size * vals_per_byte, dtype=np.uint8)
    for i in range(vals_per_byte):
        out[i::vals_per_byte] = (raw >> (nbit * i)) & mask
    if data.dtype == np.int8:
        # sign-extend nbit values
        shift = 8 - nbit
        out = ((out.astype(np.int8) << shift) >> shift).astype(np.int8)
    if data.ndim > 1:
        return out.reshape(data.shape[:-1] + (-1,))
    return out


class SigprocFile(object):
    """Streaming reader (and simple writer) for sigproc files.

    Read side mirrors the reference sigproc2.SigprocFile: `header`,
    `frame_shape` (nifs, nchans), `read(nframe)` returning unpacked
    (nframe, nifs, nchans) arrays, `readinto(buf)` for raw bytes.
    """

    def __init__(self, filename=None):
        self.f = None
        self.header = {}
        if filename is not None:
            self.open(filename)

    def open(self, filename, mode="rb"):
        if "b" not in mode:
            raise NotImplementedError("No support for non-binary files")
        self.f = open(filename, mode)
        if "r" in mode:
            self.read_header()
        return self

    def read_header(self):
        self.header = read_header(self.f)
        self.nbit = self.header["nbits"]
        signed = bool(self.header.get("signed", False))
        if self.nbit >= 8:
            if signed:
                self.dtype = {8: np.int8, 16: np.int16, 32: np.float32,
                              64: np.float64}[self.nbit]
            else:
                self.dtype = {8: np.uint8, 16: np.uint16, 32: np.float32,
                              64: np.float64}[self.nbit]
        else:
            self.dtype = np.int8 if signed else np.uint8
        self.frame_shape = (self.header["nifs"], self.header["nchans"])
        self.frame_size = self.frame_shape[0] * self.frame_shape[1]
        self.frame_nbit = self.frame_size * self.nbit
        self.frame_nbyte = self.frame_nbit // 8

    def close(self):
        if self.f is not None:
            self.f.close()
            self.f = None

    def __enter__(self):
        return self

    def __exit__(self, t, v, tb):
        self.close()

    def nframe(self):
        cur = self.f.tell()
        self.f.seek(0, os.SEEK_END)
        n = (self.f.tell() - self.header["header_size"]) * 8 \
            // self.frame_nbit
        self.f.seek(cur)
        return n

    def read(self, nframe):
        """Read up to nframe frames from the current position, unpacking
        sub-byte data to 8 bits; returns (n, nifs, nchans)."""
        nbyte = nframe * self.frame_nbit // 8
        raw = np.fromfile(self.f, count=nbyte, dtype=np.uint8)
        nframe_read = raw.size * 8 // self.frame_nbit
        raw = raw[:nframe_read * self.frame_nbit // 8]
        if self.nbit < 8:
            signed = bool(self.header.get("signed", False))
            data = unpack(raw.view(np.int8 if signed else np.uint8),
                          self.nbit)
        else:
            data = raw.view(self.dtype)
        return data.reshape((nframe_read,) + self.frame_shape)

    def readinto(self, buf):
        return self.f.readinto(buf)
