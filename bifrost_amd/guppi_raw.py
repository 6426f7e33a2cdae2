"""GUPPI RAW format header parsing (reference python/bifrost/guppi_raw.py
surface and semantics).

Format: headers are a run of 80-char space-padded records "KEY     = value",
terminated by an 'END' record; string values are quoted; if DIRECTIO is
present and non-zero the header is padded to a 512-byte boundary.  The
binary block that follows is [chan][time][pol][complex] of BLOCSIZE bytes,
with NTIME = BLOCSIZE*8 // (OBSNCHAN * NPOL * 2 * NBITS).
"""

__all__ = ["read_header"]

_RECORD_LEN = 80
_DIRECTIO_ALIGN = 512


def read_header(f):
    """Parse one GUPPI RAW block header from binary stream `f`.

    Returns a dict of key -> int/float/str.  Reads (never seeks) so Unix
    pipes work; leaves `f` positioned at the start of the binary block.
    """
    hdr = {}
    while True:
        record = f.read(_RECORD_LEN)
        if len(record) < _RECORD_LEN:
            raise IOError("EOF reached in middle of header")
        record = record.decode()
        if record.startswith("END"):
            break
        key, val = record.split("=", 1)
        key, val = key.strip(), val.strip()
        if key in hdr:
            raise KeyError("Duplicate header key: %s" % key)
        try:
            val = int(val)
        except ValueError:
            try:
                val = float(val)
            except ValueError:
                if val[:1] not in ("'", '"'):
                    raise ValueError("Invalid header value: %r" % (val,))
                val = val[1:-1].rstrip()  # unquote, drop in-string padding
        hdr[key] = val
    if "DIRECTIO" in hdr and hdr["DIRECTIO"]:
        # header padded to a 512-byte boundary; read (not seek) past it
        f.read(_DIRECTIO_ALIGN - f.tell() % _DIRECTIO_ALIGN)
    if "NPOL" in hdr:
        # files with NPOL=4 count the complex components as pols
        hdr["NPOL"] = 1 if hdr["NPOL"] == 1 else 2
    if "NTIME" not in hdr:
        hdr["NTIME"] = hdr["BLOCSIZE"] * 8 // (hdr["OBSNCHAN"] *
                                               hdr["NPOL"] * 2 *
                                               hdr["NBITS"])
    return hdr
