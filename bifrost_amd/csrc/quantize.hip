// bifrost_amd: bfQuantize — f32 -> (complex) integer with sub-byte packing
// (feeder a5).  Behaviour contract: reference
// src/quantize.cpp:230-470 semantics (rint = round-half-even, symmetric
// clipping, float math for 8/16-bit outputs, double math for 32-bit), and
// the GPU ci4 semantics (clip in float then rint, guantize.cu:52) with re
// in the HIGH nibble (Complex<FourBit>, src/Complex.hpp:149-168).
// Implementation is our own: one grid-stride streaming kernel per output
// width; HBM-bound.

#include <bifrost/quantize.h>

#include <hip/hip_runtime.h>

#include <cmath>

#include "dtype.hpp"
#include "hipctx.hpp"
#include "status.hpp"

namespace {

template <typename T>
__host__ __device__ inline T bswap(T v);
template <> __host__ __device__ inline float bswap(float v) {
    unsigned u; __builtin_memcpy(&u, &v, 4); u = __builtin_bswap32(u);
    float f; __builtin_memcpy(&f, &u, 4); return f;
}
template <> __host__ __device__ inline signed char bswap(signed char v) { return v; }
template <> __host__ __device__ inline unsigned char bswap(unsigned char v) { return v; }
template <> __host__ __device__ inline short bswap(short v) { return (short)__builtin_bswap16((unsigned short)v); }
template <> __host__ __device__ inline unsigned short bswap(unsigned short v) { return __builtin_bswap16(v); }
template <> __host__ __device__ inline int bswap(int v) { return (int)__builtin_bswap32((unsigned)v); }
template <> __host__ __device__ inline unsigned bswap(unsigned v) { return __builtin_bswap32(v); }

// One quantized value.  SType float for <=16-bit outs, double for 32-bit
// (reference quantize.cpp:392-423 type choices).
template <typename O, typename S>
__host__ __device__ inline O quant1(float x, S scale, S lo, S hi) {
    S v = (S)x * scale;
    v = v < lo ? lo : (v > hi ? hi : v);
    return (O)rint(v);
}

// 4-wide variant: float4 reads and one packed store per thread (the
// scalar kernel's 1-2 B stores bound it well under the HBM stream rate).
template <typename O, typename S>
__global__ __launch_bounds__(256) void quantize_v4_kernel(
    const float* __restrict__ in, O* __restrict__ out, size_t n4,
    S scale, S lo, S hi) {
    typedef float f4 __attribute__((ext_vector_type(4)));
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n4; i += stride) {
        f4 x = ((const f4*)in)[i];
        O q[4];
        for (int e = 0; e < 4; ++e)
            q[e] = quant1<O, S>(x[e], scale, lo, hi);
        __builtin_memcpy(&out[4 * i], q, 4 * sizeof(O));
    }
}

template <typename O, typename S, bool BSI, bool BSO>
__global__ void quantize_kernel(const float* __restrict__ in,
                                O* __restrict__ out, size_t n,
                                S scale, S lo, S hi) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        float x = in[i];
        if (BSI) x = bswap(x);
        O q = quant1<O, S>(x, scale, lo, hi);
        if (BSO) q = bswap(q);
        out[i] = q;
    }
}

// ci4: two floats -> one byte, re in HIGH nibble.
template <bool BSI>
__global__ void quantize_ci4_kernel(const float* __restrict__ in,
                                    unsigned char* __restrict__ out,
                                    size_t npair, float scale) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < npair; i += stride) {
        float re = in[2 * i], im = in[2 * i + 1];
        if (BSI) { re = bswap(re); im = bswap(im); }
        int qr = (int)quant1<int, float>(re, scale, -7.f, 7.f);
        int qi = (int)quant1<int, float>(im, scale, -7.f, 7.f);
        out[i] = (unsigned char)(((qr & 0xF) << 4) | (qi & 0xF));
    }
}

template <typename O, typename S, bool BSI, bool BSO>
void quantize_cpu(const float* in, O* out, size_t n, S scale, S lo, S hi) {
    for (size_t i = 0; i < n; ++i) {
        float x = in[i];
        if (BSI) x = bswap(x);
        O q = quant1<O, S>(x, scale, lo, hi);
        if (BSO) q = bswap(q);
        out[i] = q;
    }
}

template <bool BSI>
void quantize_ci4_cpu(const float* in, unsigned char* out, size_t npair,
                      float scale) {
    for (size_t i = 0; i < npair; ++i) {
        float re = in[2 * i], im = in[2 * i + 1];
        if (BSI) { re = bswap(re); im = bswap(im); }
        int qr = (int)quant1<int, float>(re, scale, -7.f, 7.f);
        int qi = (int)quant1<int, float>(im, scale, -7.f, 7.f);
        out[i] = (unsigned char)(((qr & 0xF) << 4) | (qi & 0xF));
    }
}

unsigned grid_for(size_t n) {
    size_t b = (n + 255) / 256;
    return (unsigned)(b < 32768 ? b : 32768);
}

template <typename O, typename S>
BFstatus run_quantize(const BFarray* in, const BFarray* out, size_t n,
                      double scale_d, S lo, S hi, bool bsi, bool bso) {
    S scale = (S)scale_d;
    bool on_gpu = bfamd::space_on_device(in->space) ||
                  bfamd::space_on_device(out->space);
    if (on_gpu) {
        BF_ASSERT(bfamd::space_device_accessible(in->space) &&
                  bfamd::space_device_accessible(out->space),
                  BF_STATUS_UNSUPPORTED_SPACE);
        hipStream_t s = bfamd::thread_stream();
        dim3 g(grid_for(n)), b(256);
        if (!bsi && !bso && n % 4 == 0 &&
            (uintptr_t)in->data % 16 == 0 &&
            (uintptr_t)out->data % (4 * sizeof(O)) == 0)
            hipLaunchKernelGGL((quantize_v4_kernel<O, S>),
                               dim3(grid_for(n / 4)), b, 0, s,
                               (const float*)in->data, (O*)out->data, n / 4,
                               scale, lo, hi);
        else if (!bsi && !bso)
            hipLaunchKernelGGL((quantize_kernel<O, S, false, false>), g, b, 0, s,
                               (const float*)in->data, (O*)out->data, n, scale, lo, hi);
        else if (bsi && !bso)
            hipLaunchKernelGGL((quantize_kernel<O, S, true, false>), g, b, 0, s,
                               (const float*)in->data, (O*)out->data, n, scale, lo, hi);
        else if (!bsi && bso)
            hipLaunchKernelGGL((quantize_kernel<O, S, false, true>), g, b, 0, s,
                               (const float*)in->data, (O*)out->data, n, scale, lo, hi);
        else
            hipLaunchKernelGGL((quantize_kernel<O, S, true, true>), g, b, 0, s,
                               (const float*)in->data, (O*)out->data, n, scale, lo, hi);
        BF_CHECK_HIP(hipGetLastError());
        return BF_STATUS_SUCCESS;
    }
    if (!bsi && !bso)
        quantize_cpu<O, S, false, false>((const float*)in->data, (O*)out->data, n, scale, lo, hi);
    else if (bsi && !bso)
        quantize_cpu<O, S, true, false>((const float*)in->data, (O*)out->data, n, scale, lo, hi);
    else if (!bsi && bso)
        quantize_cpu<O, S, false, true>((const float*)in->data, (O*)out->data, n, scale, lo, hi);
    else
        quantize_cpu<O, S, true, true>((const float*)in->data, (O*)out->data, n, scale, lo, hi);
    return BF_STATUS_SUCCESS;
}

}  // namespace

extern "C" BFstatus bfQuantize(BFarray const* in, BFarray const* out,
                               double scale) {
    using namespace bfamd;
    BF_ASSERT(in && out, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(!out->immutable, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(shapes_equal(in, out), BF_STATUS_INVALID_SHAPE);
    BF_ASSERT(dtype_is_complex(in->dtype) == dtype_is_complex(out->dtype),
              BF_STATUS_INVALID_DTYPE);
    BF_ASSERT(!dtype_is_complex(in->dtype) ||
              in->conjugated == out->conjugated, BF_STATUS_UNSUPPORTED);
    BF_ASSERT(is_contiguous(in), BF_STATUS_UNSUPPORTED_STRIDE);
    BF_ASSERT(is_contiguous(out), BF_STATUS_UNSUPPORTED_STRIDE);
    BF_ASSERT(in->dtype == BF_DTYPE_F32 || in->dtype == BF_DTYPE_CF32,
              BF_STATUS_UNSUPPORTED_DTYPE);

    size_t n = num_contiguous_elements(in);  // complex elements count once
    bool bsi = in->big_endian != 0;
    bool bso = out->big_endian != 0;
    size_t nreal = dtype_is_complex(in->dtype) ? n * 2 : n;

    switch (out->dtype & ~BF_DTYPE_COMPLEX_BIT) {
        case BF_DTYPE_I8:
            return run_quantize<signed char, float>(in, out, nreal, scale,
                                                    -127.f, 127.f, bsi, bso);
        case BF_DTYPE_I16:
            return run_quantize<short, float>(in, out, nreal, scale,
                                              -32767.f, 32767.f, bsi, bso);
        case BF_DTYPE_I32:
            return run_quantize<int, double>(in, out, nreal, scale,
                                             -2147483647.0, 2147483647.0,
                                             bsi, bso);
        case BF_DTYPE_U8 & ~BF_DTYPE_COMPLEX_BIT:
            return run_quantize<unsigned char, float>(in, out, nreal, scale,
                                                      0.f, 255.f, bsi, bso);
        case BF_DTYPE_U16 & ~BF_DTYPE_COMPLEX_BIT:
            return run_quantize<unsigned short, float>(in, out, nreal, scale,
                                                       0.f, 65535.f, bsi, bso);
        case BF_DTYPE_U32 & ~BF_DTYPE_COMPLEX_BIT:
            return run_quantize<unsigned, double>(in, out, nreal, scale, 0.0,
                                                  4294967295.0, bsi, bso);
        case BF_DTYPE_I4: {
            BF_ASSERT(out->dtype == BF_DTYPE_CI4, BF_STATUS_UNSUPPORTED_DTYPE);
            size_t npair = n;
            bool on_gpu = bfamd::space_on_device(in->space) ||
                          bfamd::space_on_device(out->space);
            if (on_gpu) {
                BF_ASSERT(bfamd::space_device_accessible(in->space) &&
                          bfamd::space_device_accessible(out->space),
                          BF_STATUS_UNSUPPORTED_SPACE);
                hipStream_t s = bfamd::thread_stream();
                dim3 g(grid_for(npair)), b(256);
                if (bsi)
                    hipLaunchKernelGGL(quantize_ci4_kernel<true>, g, b, 0, s,
                                       (const float*)in->data,
                                       (unsigned char*)out->data, npair,
                                       (float)scale);
                else
                    hipLaunchKernelGGL(quantize_ci4_kernel<false>, g, b, 0, s,
                                       (const float*)in->data,
                                       (unsigned char*)out->data, npair,
                                       (float)scale);
                BF_CHECK_HIP(hipGetLastError());
            } else if (bsi) {
                quantize_ci4_cpu<true>((const float*)in->data,
                                       (unsigned char*)out->data, npair,
                                       (float)scale);
            } else {
                quantize_ci4_cpu<false>((const float*)in->data,
                                        (unsigned char*)out->data, npair,
                                        (float)scale);
            }
            return BF_STATUS_SUCCESS;
        }
        default:
            return BF_STATUS_UNSUPPORTED_DTYPE;
    }
}
