// bifrost_amd: bfReduce — axis reductions / scrunching (SURVEY.md §8f
// row n3).  Behaviour contract: reference src/reduce.cu:880-919 —
//   * the reduced axis is inferred from the shapes (exactly one dim with
//     out->shape[d] < in->shape[d]; the factor n = in/out must divide);
//   * real inputs i8/i16/u8/u16/f32 accumulate in f32 and write f32;
//   * complex inputs ci8/ci16/cf32 write cf32 for sum/mean/stderr and
//     f32 for the power ops (|x|^2); min/max of complex is unsupported;
//   * mean = sum/n, stderr = sum/sqrt(n) (reference reduce.cu:108-114);
//   * arbitrary element-aligned strides (sliced views) on both sides.
// Implementation is our own, two kernels:
//   1. thread kernel — one thread per output element, grid-strided, inner
//      loop over the n reduced elements.  Coalesced whenever the reduced
//      axis is not the fastest (the fastest output dim maps to lane id).
//   2. wave kernel — reduced axis IS the fastest input axis and n >= 64:
//      one wave64 per output element, lanes stride the contiguous run
//      (coalesced), then a 6-step __shfl_xor cross-lane combine.

#include <bifrost/reduce.h>

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdint>
#include <type_traits>

#include "dtype.hpp"
#include "hipctx.hpp"
#include "status.hpp"

namespace {

enum OpKind { OP_SUM, OP_MIN, OP_MAX };

struct ReduceArgs {
    int ndim;                    // output ndim
    long oshape[BF_MAX_DIMS];    // output shape
    long ostrides[BF_MAX_DIMS];  // output element strides
    long istrides[BF_MAX_DIMS];  // input element stride per output step
    long n;                      // reduce factor
    long istride_r;              // input element stride of one reduced step
    float scale;                 // 1, 1/n, or 1/sqrt(n)
};

// ---- accumulator plumbing --------------------------------------------------

__device__ __forceinline__ float op_combine_sum(float a, float b) { return a + b; }

template <int OPK>
__device__ __forceinline__ float op_combine(float a, float b) {
    if (OPK == OP_MIN) return fminf(a, b);
    if (OPK == OP_MAX) return fmaxf(a, b);
    return a + b;
}

template <int OPK>
__device__ __forceinline__ float op_identity() {
    if (OPK == OP_MIN) return __builtin_inff();
    if (OPK == OP_MAX) return -__builtin_inff();
    return 0.0f;
}

struct cfloat { float re, im; };

// value loaders: real T -> float; complex pair -> cfloat
template <typename T>
__device__ __forceinline__ float load_real(const T* p, long off) {
    return (float)p[off];
}
template <typename T>
__device__ __forceinline__ cfloat load_cplx(const T* p, long off) {
    return cfloat{(float)p[2 * off], (float)p[2 * off + 1]};
}

// ---- kernels ---------------------------------------------------------------

// Decode output linear index -> (input offset, output offset), last dim
// fastest so contiguous outputs map to consecutive threads.
__device__ __forceinline__ void decode(const ReduceArgs& a, size_t i,
                                       long* ioff, long* ooff) {
    size_t rem = i;
    long io = 0, oo = 0;
    for (int d = a.ndim - 1; d >= 0; --d) {
        long idx = (long)(rem % (size_t)a.oshape[d]);
        rem /= (size_t)a.oshape[d];
        io += idx * a.istrides[d];
        oo += idx * a.ostrides[d];
    }
    *ioff = io;
    *ooff = oo;
}

// REAL path.  POWER => accumulate x*x (reference reduce.cu:84-101).
// VEC4: contiguous f32 runs with n%4==0 use float4 loads (4x fewer
// memory instructions in the short-run thread-kernel regime).
template <typename IT, int OPK, bool POWER, bool VEC4 = false>
__global__ __launch_bounds__(256) void reduce_real_thread_kernel(
        const IT* __restrict__ in, float* __restrict__ out,
        ReduceArgs args, size_t nout) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < nout; i += stride) {
        long ioff, ooff;
        decode(args, i, &ioff, &ooff);
        float acc = op_identity<OPK>();
        if (VEC4) {
            typedef float v4 __attribute__((ext_vector_type(4)));
            const v4* p = (const v4*)((const float*)in + ioff);
            for (long j = 0; j < args.n / 4; ++j) {
                v4 v = p[j];
                for (int e = 0; e < 4; ++e) {
                    float x = v[e];
                    if (POWER) x *= x;
                    acc = op_combine<OPK>(acc, x);
                }
            }
        } else {
            for (long j = 0; j < args.n; ++j) {
                float v = load_real(in, ioff + j * args.istride_r);
                if (POWER) v *= v;
                acc = op_combine<OPK>(acc, v);
            }
        }
        out[ooff] = acc * args.scale;
    }
}

template <typename IT, int OPK, bool POWER>
__global__ __launch_bounds__(256) void reduce_real_wave_kernel(
        const IT* __restrict__ in, float* __restrict__ out,
        ReduceArgs args, size_t nout) {
    int lane = threadIdx.x & 63;
    size_t w = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    size_t wstride = ((size_t)gridDim.x * blockDim.x) >> 6;
    for (; w < nout; w += wstride) {
        long ioff, ooff;
        decode(args, w, &ioff, &ooff);
        float acc = op_identity<OPK>();
        for (long j = lane; j < args.n; j += 64) {
            float v = load_real(in, ioff + j * args.istride_r);
            if (POWER) v *= v;
            acc = op_combine<OPK>(acc, v);
        }
        for (int s = 32; s >= 1; s >>= 1)
            acc = op_combine<OPK>(acc, __shfl_xor(acc, s, 64));
        if (lane == 0) out[ooff] = acc * args.scale;
    }
}

// COMPLEX standard path (sum/mean/stderr only): component-wise sums.
template <typename IT>
__global__ __launch_bounds__(256) void reduce_cplx_thread_kernel(
        const IT* __restrict__ in, float* __restrict__ out,
        ReduceArgs args, size_t nout) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < nout; i += stride) {
        long ioff, ooff;
        decode(args, i, &ioff, &ooff);
        float re = 0.0f, im = 0.0f;
        for (long j = 0; j < args.n; ++j) {
            cfloat v = load_cplx(in, ioff + j * args.istride_r);
            re += v.re;
            im += v.im;
        }
        out[2 * ooff] = re * args.scale;
        out[2 * ooff + 1] = im * args.scale;
    }
}

template <typename IT>
__global__ __launch_bounds__(256) void reduce_cplx_wave_kernel(
        const IT* __restrict__ in, float* __restrict__ out,
        ReduceArgs args, size_t nout) {
    int lane = threadIdx.x & 63;
    size_t w = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    size_t wstride = ((size_t)gridDim.x * blockDim.x) >> 6;
    for (; w < nout; w += wstride) {
        long ioff, ooff;
        decode(args, w, &ioff, &ooff);
        float re = 0.0f, im = 0.0f;
        for (long j = lane; j < args.n; j += 64) {
            cfloat v = load_cplx(in, ioff + j * args.istride_r);
            re += v.re;
            im += v.im;
        }
        for (int s = 32; s >= 1; s >>= 1) {
            re += __shfl_xor(re, s, 64);
            im += __shfl_xor(im, s, 64);
        }
        if (lane == 0) {
            out[2 * ooff] = re * args.scale;
            out[2 * ooff + 1] = im * args.scale;
        }
    }
}

// COMPLEX power path: |x|^2 then a real reduction.
template <typename IT, int OPK>
__global__ __launch_bounds__(256) void reduce_cpow_thread_kernel(
        const IT* __restrict__ in, float* __restrict__ out,
        ReduceArgs args, size_t nout) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < nout; i += stride) {
        long ioff, ooff;
        decode(args, i, &ioff, &ooff);
        float acc = op_identity<OPK>();
        for (long j = 0; j < args.n; ++j) {
            cfloat v = load_cplx(in, ioff + j * args.istride_r);
            acc = op_combine<OPK>(acc, v.re * v.re + v.im * v.im);
        }
        out[ooff] = acc * args.scale;
    }
}

template <typename IT, int OPK>
__global__ __launch_bounds__(256) void reduce_cpow_wave_kernel(
        const IT* __restrict__ in, float* __restrict__ out,
        ReduceArgs args, size_t nout) {
    int lane = threadIdx.x & 63;
    size_t w = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    size_t wstride = ((size_t)gridDim.x * blockDim.x) >> 6;
    for (; w < nout; w += wstride) {
        long ioff, ooff;
        decode(args, w, &ioff, &ooff);
        float acc = op_identity<OPK>();
        for (long j = lane; j < args.n; j += 64) {
            cfloat v = load_cplx(in, ioff + j * args.istride_r);
            acc = op_combine<OPK>(acc, v.re * v.re + v.im * v.im);
        }
        for (int s = 32; s >= 1; s >>= 1)
            acc = op_combine<OPK>(acc, __shfl_xor(acc, s, 64));
        if (lane == 0) out[ooff] = acc * args.scale;
    }
}

// ---- launch plumbing -------------------------------------------------------

// The VEC4 fast path does 16-B dwordx4 loads at in + ioff; a base pointer
// check alone is not enough — per-output offsets built from sliced/padded
// outer-dim strides can break the 16-B alignment.  Require every outer
// stride to be a multiple of 4 elements.
inline bool vec4_strides_ok(const ReduceArgs& args) {
    for (int d = 0; d < args.ndim; ++d)
        if (args.istrides[d] % 4 != 0) return false;
    return true;
}

inline int thread_blocks(size_t nout) {
    size_t b = (nout + 255) / 256;
    if (b > 16384) b = 16384;  // grid-strided beyond this (>> 256 WGs)
    return (int)b;
}

inline int wave_blocks(size_t nout) {
    size_t b = (nout + 3) / 4;  // 4 waves (outputs) per 256-thread block
    if (b > 16384) b = 16384;
    return (int)b;
}

template <typename IT, int OPK, bool POWER>
void launch_real(const void* in, void* out, const ReduceArgs& args,
                 size_t nout, bool wave, hipStream_t s) {
    if (wave)
        hipLaunchKernelGGL((reduce_real_wave_kernel<IT, OPK, POWER>),
                           dim3(wave_blocks(nout)), dim3(256), 0, s,
                           (const IT*)in, (float*)out, args, nout);
    else if (std::is_same<IT, float>::value && args.istride_r == 1 &&
             args.n % 4 == 0 && ((uintptr_t)in % 16 == 0) &&
             vec4_strides_ok(args))
        hipLaunchKernelGGL(
            (reduce_real_thread_kernel<IT, OPK, POWER, true>),
            dim3(thread_blocks(nout)), dim3(256), 0, s, (const IT*)in,
            (float*)out, args, nout);
    else
        hipLaunchKernelGGL((reduce_real_thread_kernel<IT, OPK, POWER>),
                           dim3(thread_blocks(nout)), dim3(256), 0, s,
                           (const IT*)in, (float*)out, args, nout);
}

template <typename IT>
void launch_cplx(const void* in, void* out, const ReduceArgs& args,
                 size_t nout, bool wave, hipStream_t s) {
    if (wave)
        hipLaunchKernelGGL((reduce_cplx_wave_kernel<IT>),
                           dim3(wave_blocks(nout)), dim3(256), 0, s,
                           (const IT*)in, (float*)out, args, nout);
    else
        hipLaunchKernelGGL((reduce_cplx_thread_kernel<IT>),
                           dim3(thread_blocks(nout)), dim3(256), 0, s,
                           (const IT*)in, (float*)out, args, nout);
}

template <typename IT, int OPK>
void launch_cpow(const void* in, void* out, const ReduceArgs& args,
                 size_t nout, bool wave, hipStream_t s) {
    if (wave)
        hipLaunchKernelGGL((reduce_cpow_wave_kernel<IT, OPK>),
                           dim3(wave_blocks(nout)), dim3(256), 0, s,
                           (const IT*)in, (float*)out, args, nout);
    else
        hipLaunchKernelGGL((reduce_cpow_thread_kernel<IT, OPK>),
                           dim3(thread_blocks(nout)), dim3(256), 0, s,
                           (const IT*)in, (float*)out, args, nout);
}

template <typename IT>
BFstatus dispatch_real(const BFarray* in, const BFarray* out,
                       BFreduce_op op, const ReduceArgs& args, size_t nout,
                       bool wave, hipStream_t s) {
    switch (op) {
    case BF_REDUCE_SUM:
    case BF_REDUCE_MEAN:
    case BF_REDUCE_STDERR:
        launch_real<IT, OP_SUM, false>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_MIN:
        launch_real<IT, OP_MIN, false>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_MAX:
        launch_real<IT, OP_MAX, false>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_POWER_SUM:
    case BF_REDUCE_POWER_MEAN:
    case BF_REDUCE_POWER_STDERR:
        launch_real<IT, OP_SUM, true>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_POWER_MIN:
        launch_real<IT, OP_MIN, true>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_POWER_MAX:
        launch_real<IT, OP_MAX, true>(in->data, out->data, args, nout, wave, s);
        break;
    default:
        return BF_STATUS_INVALID_ARGUMENT;
    }
    BF_CHECK_HIP(hipGetLastError());
    return BF_STATUS_SUCCESS;
}

template <typename IT>
BFstatus dispatch_cplx(const BFarray* in, const BFarray* out,
                       BFreduce_op op, const ReduceArgs& args, size_t nout,
                       bool wave, hipStream_t s) {
    switch (op) {
    case BF_REDUCE_SUM:
    case BF_REDUCE_MEAN:
    case BF_REDUCE_STDERR:
        launch_cplx<IT>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_POWER_SUM:
    case BF_REDUCE_POWER_MEAN:
    case BF_REDUCE_POWER_STDERR:
        launch_cpow<IT, OP_SUM>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_POWER_MIN:
        launch_cpow<IT, OP_MIN>(in->data, out->data, args, nout, wave, s);
        break;
    case BF_REDUCE_POWER_MAX:
        launch_cpow<IT, OP_MAX>(in->data, out->data, args, nout, wave, s);
        break;
    default:
        // min/max of a complex value is undefined (reference docstring,
        // blocks/reduce.py:104-105)
        return BF_STATUS_UNSUPPORTED;
    }
    BF_CHECK_HIP(hipGetLastError());
    return BF_STATUS_SUCCESS;
}

}  // namespace

extern "C" BFstatus bfReduce(BFarray const* in, BFarray const* out,
                             BFreduce_op op) {
    using namespace bfamd;
    BF_ASSERT(in && out, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(in->data && out->data, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(in->ndim == out->ndim, BF_STATUS_INVALID_SHAPE);
    BF_ASSERT(space_device_accessible(in->space) &&
              space_device_accessible(out->space),
              BF_STATUS_UNSUPPORTED_SPACE);
    BF_ASSERT(op >= BF_REDUCE_SUM && op <= BF_REDUCE_POWER_STDERR,
              BF_STATUS_INVALID_ARGUMENT);

    // Infer the reduced axis (reference src/reduce.cu:904-916): exactly one.
    int ndim = in->ndim;
    int rdim = -1;
    for (int d = 0; d < ndim; ++d) {
        BF_ASSERT(out->shape[d] <= in->shape[d], BF_STATUS_INVALID_SHAPE);
        if (out->shape[d] < in->shape[d]) {
            BF_ASSERT(rdim < 0, BF_STATUS_UNSUPPORTED_SHAPE);
            rdim = d;
        }
    }
    BF_ASSERT(rdim >= 0, BF_STATUS_INVALID_SHAPE);
    BF_ASSERT(in->shape[rdim] % out->shape[rdim] == 0,
              BF_STATUS_INVALID_SHAPE);
    long n = in->shape[rdim] / out->shape[rdim];

    bool cplx = dtype_is_complex(in->dtype);
    bool power = op >= BF_REDUCE_POWER_SUM;
    // output dtype contract (reference reduce.cu:349-366, 851-876)
    if (cplx && !power) {
        BF_ASSERT(out->dtype == BF_DTYPE_CF32, BF_STATUS_UNSUPPORTED_DTYPE);
    } else {
        BF_ASSERT(out->dtype == BF_DTYPE_F32, BF_STATUS_UNSUPPORTED_DTYPE);
    }

    // element-aligned strides on both sides
    int inb = dtype_nbyte(in->dtype);
    int onb = dtype_nbyte(out->dtype);
    ReduceArgs args;
    args.ndim = ndim;
    args.n = n;
    size_t nout = 1;
    for (int d = 0; d < ndim; ++d) {
        BF_ASSERT(in->strides[d] % inb == 0, BF_STATUS_UNSUPPORTED_STRIDE);
        BF_ASSERT(out->strides[d] % onb == 0, BF_STATUS_UNSUPPORTED_STRIDE);
        args.oshape[d] = out->shape[d];
        args.ostrides[d] = out->strides[d] / onb;
        long istride_elem = in->strides[d] / inb;
        args.istrides[d] = (d == rdim) ? istride_elem * n : istride_elem;
        nout *= (size_t)out->shape[d];
    }
    args.istride_r = in->strides[rdim] / inb;
    switch (op) {
    case BF_REDUCE_MEAN:
    case BF_REDUCE_POWER_MEAN:
        args.scale = 1.0f / (float)n;
        break;
    case BF_REDUCE_STDERR:
    case BF_REDUCE_POWER_STDERR:
        args.scale = 1.0f / std::sqrt((float)n);
        break;
    default:
        args.scale = 1.0f;
    }
    if (nout == 0) return BF_STATUS_SUCCESS;

    // wave kernel pays when the reduced run is contiguous and long
    bool wave = (args.istride_r == 1) && (n >= 64);
    hipStream_t s = thread_stream();

    switch (in->dtype) {
    case BF_DTYPE_I8:
        return dispatch_real<int8_t>(in, out, op, args, nout, wave, s);
    case BF_DTYPE_I16:
        return dispatch_real<int16_t>(in, out, op, args, nout, wave, s);
    case BF_DTYPE_U8:
        return dispatch_real<uint8_t>(in, out, op, args, nout, wave, s);
    case BF_DTYPE_U16:
        return dispatch_real<uint16_t>(in, out, op, args, nout, wave, s);
    case BF_DTYPE_F32:
        return dispatch_real<float>(in, out, op, args, nout, wave, s);
    case BF_DTYPE_CI8:
        return dispatch_cplx<int8_t>(in, out, op, args, nout, wave, s);
    case BF_DTYPE_CI16:
        return dispatch_cplx<int16_t>(in, out, op, args, nout, wave, s);
    case BF_DTYPE_CF32:
        return dispatch_cplx<float>(in, out, op, args, nout, wave, s);
    default:
        return BF_STATUS_UNSUPPORTED_DTYPE;
    }
}
