"""Bit-exact numpy restatement of bfUnpack / bfQuantize — TEST ORACLE ONLY.

Restates, element for element, the CPU bit-twiddle logic of the reference:
  - unpack:   /root/reference/src/unpack.cpp:41-224 (the unpack() overloads
              for 1/2/4-bit signed/unsigned and the foreach_simple/promote
              drivers, dispatch at unpack.cpp:242-534)
  - quantize: /root/reference/src/quantize.cpp:51-240 (quantize(),
              foreach_simple_cpu_{4bit,2bit,1bit}, dispatch :230-470)

Pinned by the reference's own known-answer vectors (test/test_unpack.py:33-95,
test/test_quantize.py:33-50) — see tests/test_oracle_bitops.py — and
cross-validatable against the reference CPU code itself compiled by
oracle/ref_build/Makefile (oracle.refcpu).

Conventions captured here (verified against the vectors):
  - Packed sub-byte elements unpack LSB-first: for ci4 byte 0xXY the REAL
    part is the LOW nibble Y and the IMAG part the HIGH nibble X
    (0x10 -> (re=0, im=1), test_unpack.py:41-46).
  - bfQuantize's ci4 packing is the OPPOSITE nibble order (re in the HIGH
    nibble, quantize.cpp:137-138, matching Complex<FourBit> re<<4|im in
    src/Complex.hpp) — the reference is internally inconsistent here and we
    preserve each op's behaviour exactly.
  - Rounding is rint = round-half-to-even (0.5 -> 0, 1.5 -> 2).
  - Signed integer clipping is symmetric: +-127 for i8 (never -128),
    +-7 for the 4-bit path (quantize.cpp:46-67).
  - The reference's 1-bit quantize drops inputs A,B,C of each group of 8
    (masks at quantize.cpp:216-223); restated verbatim.
  - ci4 quantize rounding follows the reference's GPU path (clip in float
    then rint, guantize.cu:52,157-158); the reference's CPU path instead
    truncates through an int8_t-typed clip (quantize.cpp:65-67) and the two
    disagree on non-integer inputs.  Our product kernel must match the GPU
    semantics, so the oracle does too (packing pinned against the CPU build
    on integer inputs, where they agree).
  - Sub-byte dtypes other than ci4 are UNREACHABLE through the reference ABI
    (is_contiguous computes NBYTE==0 and rejects them, utils.hpp:258-269);
    their restatements here document the dead bit logic only.
"""

import numpy as np

__all__ = ["unpack", "quantize"]


# ---------------------------------------------------------------------------
# unpack
# ---------------------------------------------------------------------------

_NBIT = {"i1": 1, "i2": 2, "i4": 4, "u1": 1, "u2": 2, "u4": 4,
         "ci1": 1, "ci2": 2, "ci4": 4}


def _unpack_bytes(raw, nbit, signed, align_msb, byteswap):
    """Unpack a flat uint8 array into one int8/uint8 per nbit-wide sub-word.

    Sub-words come out LSB-first per byte (unpack.cpp:41-197); byteswap=True
    reverses the order within each input byte (the reference byteswaps the
    packed word, which for <=8-bit groups is a sub-word order reversal).
    """
    raw = np.ascontiguousarray(raw, dtype=np.uint8)
    m = 8 // nbit                      # sub-words per byte
    mask = (1 << nbit) - 1
    # shape [nbytes, m]: sub-word j of byte b at bits [nbit*j +: nbit]
    shifts = np.arange(m, dtype=np.uint8) * nbit
    sub = (raw[:, None] >> shifts[None, :]) & mask      # uint8, LSB-first
    if byteswap:
        sub = sub[:, ::-1]
    if align_msb:
        out = (sub.astype(np.uint8) << (8 - nbit)).view(np.int8 if signed else np.uint8)
    else:
        if signed:
            # sign-extend: place at MSB then arithmetic shift right
            out = (sub.astype(np.uint8) << (8 - nbit)).view(np.int8) >> (8 - nbit)
        else:
            out = sub.astype(np.uint8)
    return np.ascontiguousarray(out).reshape(-1)


def unpack(raw, in_dtype, out_dtype, align_msb=False, byteswap=False,
           conjugate=False):
    """bfUnpack oracle (unpack.cpp:242-534).

    raw:      flat uint8 array of packed input bytes.
    in_dtype: one of i1,i2,i4,u2,u4,ci1,ci2,ci4.
    out_dtype: 'i8'/'ci8' (int8 stream), 'f32'/'cf32', 'f64'/'cf64'.
    Returns the flat unpacked stream (int8/uint8 for *i8 outputs, else float).
    Complex types are simply 2x the real count; `conjugate` negates every
    second sub-word (the imaginary parts) for signed complex inputs.
    """
    nbit = _NBIT[in_dtype]
    signed = not in_dtype.startswith("u")
    complex_in = in_dtype.startswith("c")
    ints = _unpack_bytes(raw, nbit, signed, align_msb, byteswap)
    if conjugate:
        if not (complex_in and signed):
            raise ValueError("conjugate requires signed complex input")
        ints = ints.copy()
        ints[1::2] = -ints[1::2]
    if out_dtype in ("i8", "ci8", "u8"):
        return ints
    if out_dtype in ("f32", "cf32"):
        return ints.astype(np.float32)
    if out_dtype in ("f64", "cf64"):
        return ints.astype(np.float64)
    raise ValueError("unsupported unpack output dtype %r" % (out_dtype,))


# ---------------------------------------------------------------------------
# quantize
# ---------------------------------------------------------------------------

def _rint(x):
    return np.rint(x)


def quantize(data, out_dtype, scale=1.0, byteswap_in=False, byteswap_out=False):
    """bfQuantize oracle (quantize.cpp:80-240, dispatch :276-470).

    data: flat float32 array (pairs re,im for complex inputs).
    out_dtype: i8/ci8, i16/ci16, i32/ci32, u8/u16/u32, ci4, ci2, ci1.
    Returns flat int array, or packed uint8 for sub-byte outputs.
    Scale-type fidelity: float math for 8/16-bit outs, double for 32-bit
    (quantize.cpp:392-423).
    """
    x = np.asarray(data, dtype=np.float32).reshape(-1)
    if byteswap_in:
        x = x.byteswap()
    base = out_dtype.lstrip("c")
    if base in ("i8", "i16", "u8", "u16"):
        s = np.float32(scale)
        lim = {"i8": 127, "i16": 32767, "u8": 255, "u16": 65535}[base]
        lo = -lim if base.startswith("i") else 0
        q = _rint(np.clip(x * s, np.float32(lo), np.float32(lim)))
        out = q.astype({"i8": np.int8, "i16": np.int16,
                        "u8": np.uint8, "u16": np.uint16}[base])
    elif base in ("i32", "u32"):
        s = np.float64(scale)
        lim = 2147483647 if base == "i32" else 4294967295
        lo = -lim if base == "i32" else 0
        q = _rint(np.clip(x.astype(np.float64) * s, float(lo), float(lim)))
        out = q.astype(np.int32 if base == "i32" else np.uint32)
    elif base == "i4":
        # foreach_simple_cpu_4bit (quantize.cpp:124-147): pairs (re, im) ->
        # one byte, RE IN THE HIGH NIBBLE.  clip to +-7.
        s = np.float32(scale)
        q = _rint(np.clip(x * s, np.float32(-7), np.float32(7))).astype(np.int8)
        qr, qi = q[0::2].astype(np.int32), q[1::2].astype(np.int32)
        out = (((qr * 16) & 0xF0) | (((qi * 16) >> 4) & 0x0F)).astype(np.uint8)
    elif base == "i2":
        # foreach_simple_cpu_2bit (quantize.cpp:149-180): 4 values/byte,
        # first value in bits 7:6.  clip_2bit to [-1, 1].
        s = np.float32(scale)
        q = _rint(np.clip(x * s, np.float32(-1), np.float32(1))).astype(np.int32)
        a, b, c, d = q[0::4], q[1::4], q[2::4], q[3::4]
        out = ((((a * 64)) & 0xC0) | (((b * 64) >> 2) & 0x30)
               | (((c * 64) >> 4) & 0x0C) | (((d * 64) >> 6) & 0x03)).astype(np.uint8)
    elif base == "i1":
        # foreach_simple_cpu_1bit (quantize.cpp:182-240): clip_1bit maps
        # x>=0 -> 1 else 0; the reference's mask set drops values A,B,C of
        # each group of 8 and packs D..H into bits 4..0.  Restated verbatim.
        s = np.float32(scale)
        q = (x * s >= 0).astype(np.int32) * 128
        grp = [q[i::8] for i in range(8)]
        out = (((grp[0]) & 0x08) | ((grp[1] >> 1) & 0x04) | ((grp[2] >> 2) & 0x02)
               | ((grp[3] >> 3) & 0x10) | ((grp[4] >> 4) & 0x08)
               | ((grp[5] >> 5) & 0x04) | ((grp[6] >> 6) & 0x02)
               | ((grp[7] >> 7) & 0x01)).astype(np.uint8)
    else:
        raise ValueError("unsupported quantize output dtype %r" % (out_dtype,))
    if byteswap_out:
        out = out.byteswap()
    return out
