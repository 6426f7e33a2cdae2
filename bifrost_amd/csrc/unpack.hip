// bifrost_amd: bfUnpack — sub-byte sample unpacking (feeder a4 of the hot
// path).  Behaviour contract: reference src/unpack.cpp
// :242-534 + the known-answer vectors of test/test_unpack.py:33-95.
// Implementation is our own: per-byte nibble shifts (host+device shared),
// GPU kernel is a grid-stride byte stream (one ci4 byte -> one short),
// coalesced on both sides; HBM-bound.
//
// Conventions (see oracle/bitops.py):
//   ci4 byte 0xXY -> (re = sext(Y), im = sext(X))  [LSB-first sub-words]
//   byteswap (big_endian input) reverses the sub-word order;
//   align_msb keeps values shifted to the top of the byte;
//   conjugate negates the imaginary (odd) outputs.
// Only ci4 input is reachable through the ABI (is_contiguous computes
// NBYTE==0 for every other sub-byte dtype and rejects it — same as the
// reference, utils.hpp:258-269).

#include <bifrost/unpack.h>

#include <hip/hip_runtime.h>

#include "dtype.hpp"
#include "hipctx.hpp"
#include "status.hpp"

namespace {

// ci4 byte -> two int8 (re, im order already applied)
template <bool BSWAP, bool MSB, bool CONJ>
__host__ __device__ inline void unpack_ci4_byte(unsigned char b, signed char* out) {
    signed char lo, hi;
    if (MSB) {
        lo = (signed char)(b << 4);
        hi = (signed char)(b & 0xF0);
    } else {
        lo = (signed char)((signed char)(b << 4) >> 4);
        hi = (signed char)((signed char)b >> 4);
    }
    signed char re = BSWAP ? hi : lo;
    signed char im = BSWAP ? lo : hi;
    if (CONJ) im = (signed char)(-im);
    out[0] = re;
    out[1] = im;
}

template <bool BSWAP, bool MSB, bool CONJ>
__global__ void unpack_ci4_ci8_kernel(const unsigned char* __restrict__ in,
                                      short* __restrict__ out, size_t n) {
    // vectorized main body: 4 input bytes (uint) -> 8 output bytes (uint2)
    size_t n4 = n / 4;
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    const unsigned* in4 = (const unsigned*)in;
    for (size_t v = i; v < n4; v += stride) {
        unsigned d = in4[v];
        unsigned lo = 0, hi = 0;
        for (int b = 0; b < 2; ++b) {
            signed char pair[2];
            unpack_ci4_byte<BSWAP, MSB, CONJ>((unsigned char)(d >> (8 * b)),
                                              pair);
            lo |= ((unsigned)(unsigned char)pair[0] << (16 * b)) |
                  ((unsigned)(unsigned char)pair[1] << (16 * b + 8));
            unpack_ci4_byte<BSWAP, MSB, CONJ>(
                (unsigned char)(d >> (8 * (b + 2))), pair);
            hi |= ((unsigned)(unsigned char)pair[0] << (16 * b)) |
                  ((unsigned)(unsigned char)pair[1] << (16 * b + 8));
        }
        ((uint2*)out)[v] = make_uint2(lo, hi);
    }
    for (size_t t = n4 * 4 + i; t < n; t += stride) {
        signed char pair[2];
        unpack_ci4_byte<BSWAP, MSB, CONJ>(in[t], pair);
        out[t] = (short)((unsigned char)pair[0] |
                         ((unsigned short)(unsigned char)pair[1] << 8));
    }
}

template <bool BSWAP, bool MSB, bool CONJ, typename F>
__global__ void unpack_ci4_float_kernel(const unsigned char* __restrict__ in,
                                        F* __restrict__ out, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        signed char pair[2];
        unpack_ci4_byte<BSWAP, MSB, CONJ>(in[i], pair);
        out[2 * i] = (F)pair[0];
        out[2 * i + 1] = (F)pair[1];
    }
}

template <bool BSWAP, bool MSB, bool CONJ>
void unpack_ci4_cpu(const unsigned char* in, signed char* out8, float* outf,
                    double* outd, size_t n) {
    for (size_t i = 0; i < n; ++i) {
        signed char pair[2];
        unpack_ci4_byte<BSWAP, MSB, CONJ>(in[i], pair);
        if (out8) { out8[2 * i] = pair[0]; out8[2 * i + 1] = pair[1]; }
        if (outf) { outf[2 * i] = pair[0]; outf[2 * i + 1] = pair[1]; }
        if (outd) { outd[2 * i] = pair[0]; outd[2 * i + 1] = pair[1]; }
    }
}

struct LaunchCfg {
    dim3 grid, block;
};
LaunchCfg stream_cfg(size_t n) {
    // >> 256 workgroups to fill 8 XCDs x 32 CUs (MI355X); grid-stride tail.
    unsigned blocks = (unsigned)std::min<size_t>((n + 255) / 256, 32768);
    return {dim3(blocks), dim3(256)};
}

template <bool BSWAP, bool MSB, bool CONJ>
BFstatus unpack_ci4_dispatch(const BFarray* in, const BFarray* out, size_t n) {
    bool on_gpu = bfamd::space_device_accessible(in->space) &&
                  bfamd::space_on_device(out->space);
    if (on_gpu) {
        LaunchCfg cfg = stream_cfg(n);
        hipStream_t s = bfamd::thread_stream();
        if (out->dtype == BF_DTYPE_CI8 || out->dtype == BF_DTYPE_I8) {
            hipLaunchKernelGGL((unpack_ci4_ci8_kernel<BSWAP, MSB, CONJ>),
                               cfg.grid, cfg.block, 0, s,
                               (const unsigned char*)in->data,
                               (short*)out->data, n);
        } else if (out->dtype == BF_DTYPE_CF32 || out->dtype == BF_DTYPE_F32) {
            hipLaunchKernelGGL((unpack_ci4_float_kernel<BSWAP, MSB, CONJ, float>),
                               cfg.grid, cfg.block, 0, s,
                               (const unsigned char*)in->data,
                               (float*)out->data, n);
        } else if (out->dtype == BF_DTYPE_CF64 || out->dtype == BF_DTYPE_F64) {
            hipLaunchKernelGGL((unpack_ci4_float_kernel<BSWAP, MSB, CONJ, double>),
                               cfg.grid, cfg.block, 0, s,
                               (const unsigned char*)in->data,
                               (double*)out->data, n);
        } else {
            return BF_STATUS_UNSUPPORTED_DTYPE;
        }
        BF_CHECK_HIP(hipGetLastError());
        return BF_STATUS_SUCCESS;
    }
    // CPU path (reference ships one too; system-space arrays only).
    BF_ASSERT(bfamd::space_host_accessible(in->space) &&
              bfamd::space_host_accessible(out->space),
              BF_STATUS_UNSUPPORTED_SPACE);
    if (out->dtype == BF_DTYPE_CI8 || out->dtype == BF_DTYPE_I8) {
        unpack_ci4_cpu<BSWAP, MSB, CONJ>((const unsigned char*)in->data,
                                         (signed char*)out->data, nullptr,
                                         nullptr, n);
    } else if (out->dtype == BF_DTYPE_CF32 || out->dtype == BF_DTYPE_F32) {
        unpack_ci4_cpu<BSWAP, MSB, CONJ>((const unsigned char*)in->data,
                                         nullptr, (float*)out->data, nullptr, n);
    } else if (out->dtype == BF_DTYPE_CF64 || out->dtype == BF_DTYPE_F64) {
        unpack_ci4_cpu<BSWAP, MSB, CONJ>((const unsigned char*)in->data,
                                         nullptr, nullptr, (double*)out->data, n);
    } else {
        return BF_STATUS_UNSUPPORTED_DTYPE;
    }
    return BF_STATUS_SUCCESS;
}

}  // namespace

extern "C" BFstatus bfUnpack(BFarray const* in, BFarray const* out,
                             BFbool align_msb) {
    using namespace bfamd;
    BF_ASSERT(in && out, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(!out->immutable, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(shapes_equal(in, out), BF_STATUS_INVALID_SHAPE);
    BF_ASSERT(dtype_is_complex(in->dtype) == dtype_is_complex(out->dtype),
              BF_STATUS_INVALID_DTYPE);
    BF_ASSERT(dtype_is_complex(in->dtype) || !in->conjugated,
              BF_STATUS_INVALID_DTYPE);
    BF_ASSERT(is_contiguous(in), BF_STATUS_UNSUPPORTED_STRIDE);
    BF_ASSERT(is_contiguous(out), BF_STATUS_UNSUPPORTED_STRIDE);

    BF_ASSERT(in->dtype == BF_DTYPE_CI4, BF_STATUS_UNSUPPORTED_DTYPE);
    size_t nbytes = capacity_bytes(in);  // one ci4 complex per byte

    bool byteswap = in->big_endian != 0;  // we are little-endian
    bool conjugate = (in->conjugated != out->conjugated);
    bool msb = align_msb != 0;

#define DISPATCH(BS, MSBV, CJ) \
    return unpack_ci4_dispatch<BS, MSBV, CJ>(in, out, nbytes)
    if (!byteswap && !msb && !conjugate) DISPATCH(false, false, false);
    if (!byteswap && !msb && conjugate)  DISPATCH(false, false, true);
    if (!byteswap && msb && !conjugate)  DISPATCH(false, true, false);
    if (!byteswap && msb && conjugate)   DISPATCH(false, true, true);
    if (byteswap && !msb && !conjugate)  DISPATCH(true, false, false);
    if (byteswap && !msb && conjugate)   DISPATCH(true, false, true);
    if (byteswap && msb && !conjugate)   DISPATCH(true, true, false);
    DISPATCH(true, true, true);
#undef DISPATCH
}
