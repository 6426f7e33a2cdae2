// bifrost_amd: bfFft — FFTs over BFarray axes on hipFFT/rocFFT
// (SURVEY.md §8f row n2).  Behaviour contract: reference src/fft.cu +
// src/fft_kernels.cu semantics —
//   * unnormalized transforms (cuFFT convention);
//   * c2c/r2c/c2r by dtype pairing; i8/i16/u8/u16 real input converts to
//     f32 scaled by 1/(maxval+1) (fft_kernels.cu:178-191), done here by
//     an explicit conversion kernel instead of a cuFFT load callback;
//   * 1-3 transform axes in ascending order, NOT necessarily
//     consecutive: gap dims fold into the plan's inembed/onembed (their
//     strides are nested products) and are iterated as host-side batch
//     loops; one non-inner dim can ride the plan's batch stride;
//   * apply_fftshift: the SPECTRUM side is stored shifted — forward
//     transforms fftshift their output (post-pass roll), inverse/c2r
//     transforms ifftshift their input (pre-pass roll), via explicit
//     roll kernels (reference does it in load callbacks).
// Arrays must be contiguous (the reference supports strided input via
// callbacks only for some cases; we require contiguity and the Python
// layer passes contiguous arrays).

#include <bifrost/fft.h>

#include <hip/hip_runtime.h>
#include <hipfft/hipfft.h>

#include <cstring>
#include <vector>

#include "dtype.hpp"
#include "hipctx.hpp"
#include "status.hpp"

#define BF_CHECK_HIPFFT(call)                                             \
    do {                                                                  \
        hipfftResult r_ = (call);                                         \
        if (r_ != HIPFFT_SUCCESS) {                                       \
            if (bfamd::debug_enabled())                                   \
                std::fprintf(stderr, "[bifrost_amd] hipfft error %d at "  \
                             "%s:%d\n", (int)r_, __FILE__, __LINE__);     \
            return BF_STATUS_DEVICE_ERROR;                                \
        }                                                                 \
    } while (0)

namespace {

// ---- roll (fftshift / ifftshift) kernel -----------------------------------
// Gather-copy with a per-axis cyclic shift: out[..., k, ...] =
// in[..., (k + shift) % n, ...].  Element type erased to 4/8/16 bytes.
struct RollArgs {
    int ndim;
    long shape[BF_MAX_DIMS];
    long shift[BF_MAX_DIMS];  // 0 for non-transform axes
};

template <typename T>
__global__ __launch_bounds__(256) void roll_kernel(const T* __restrict__ in,
                                                   T* __restrict__ out,
                                                   RollArgs args, size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        size_t rem = i;
        size_t src = 0;
        size_t mul = 1;
        for (int d = args.ndim - 1; d >= 0; --d) {
            long nd = args.shape[d];
            long idx = (long)(rem % (size_t)nd);
            rem /= (size_t)nd;
            long sidx = idx + args.shift[d];
            if (sidx >= nd) sidx -= nd;
            src += (size_t)sidx * mul;
            mul *= (size_t)nd;
        }
        out[i] = in[src];
    }
}

// ---- integer -> float conversion kernel -----------------------------------
// reference fft_kernels.cu:178-191: result = val * (1/(maxval+1)),
// maxval = (1 << (nbit - is_signed)) - 1  =>  scale = 1/2^(nbit-is_signed).
template <typename T>
__global__ __launch_bounds__(256) void convert_kernel(
        const T* __restrict__ in, float* __restrict__ out, float scale,
        size_t n) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    size_t stride = (size_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) out[i] = (float)in[i] * scale;
}

inline int grid_for(size_t n) {
    size_t b = (n + 255) / 256;
    if (b > 16384) b = 16384;
    return (int)b;
}

}  // namespace

struct BFfft_impl {
    hipfftHandle plan = 0;
    bool have_plan = false;
    hipfftType type = HIPFFT_C2C;
    int rank = 0;
    int axlist[3] = {0};
    bool f64 = false;
    bool r2c = false, c2r = false;
    bool do_shift = false;
    // integer input conversion
    bool int_in = false;
    float int_scale = 1.0f;
    int in_elem_nbyte = 0;  // bytes per input element as stored
    // host batch loops: element strides into the PLANNED input (after
    // conversion) and the output, one entry per looped dim
    std::vector<long> loop_shape, loop_istride, loop_ostride;
    long nloop = 1;
    // spectrum roll geometry (full logical array of the spectrum side)
    RollArgs roll;
    size_t spec_elems = 0;   // elements in the spectrum-side array
    size_t in_elems = 0;     // elements in the input array
    // temp device buffers
    void* tmp_conv = nullptr;   // f32 conversion target
    void* tmp_roll = nullptr;   // roll staging (spectrum-side type)
    size_t tmp_conv_size = 0, tmp_roll_size = 0;
    int ndim = 0;
    long ishape[BF_MAX_DIMS] = {0};

    ~BFfft_impl() {
        if (have_plan) hipfftDestroy(plan);
        if (tmp_conv) hipFree(tmp_conv);
        if (tmp_roll) hipFree(tmp_roll);
    }
};

extern "C" {

BFstatus bfFftCreate(BFfft* plan_ptr) {
    BF_ASSERT(plan_ptr, BF_STATUS_INVALID_POINTER);
    BF_TRY_RETURN({ *plan_ptr = new BFfft_impl(); });
}

BFstatus bfFftDestroy(BFfft plan) {
    BF_ASSERT(plan, BF_STATUS_INVALID_HANDLE);
    delete plan;
    return BF_STATUS_SUCCESS;
}

BFstatus bfFftInit(BFfft plan, BFarray const* in, BFarray const* out,
                   int ndim, int const* axes, BFbool apply_fftshift,
                   size_t* tmp_storage_size) {
    using namespace bfamd;
    BF_ASSERT(plan && in && out && axes, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(ndim >= 1 && ndim <= 3, BF_STATUS_UNSUPPORTED_SHAPE);
    BF_ASSERT(in->ndim == out->ndim, BF_STATUS_INVALID_SHAPE);
    BF_ASSERT(space_device_accessible(in->space) &&
              space_device_accessible(out->space),
              BF_STATUS_UNSUPPORTED_SPACE);
    BF_ASSERT(is_contiguous(in) && is_contiguous(out),
              BF_STATUS_UNSUPPORTED_STRIDE);

    bool in_cplx = dtype_is_complex(in->dtype);
    bool out_cplx = dtype_is_complex(out->dtype);
    BF_ASSERT(in_cplx || out_cplx, BF_STATUS_UNSUPPORTED_DTYPE);
    int in_nbit = in->dtype & BF_DTYPE_NBIT_BITS;
    int out_nbit = out->dtype & BF_DTYPE_NBIT_BITS;

    // integer real input (i8/i16/u8/u16 -> cf32): converted to f32 scaled
    // by 1/2^(nbit-1) (signed) / 1/2^nbit (unsigned)
    plan->int_in = false;
    bool f64;
    if (!in_cplx && !dtype_is_float(in->dtype)) {
        BF_ASSERT(in_nbit == 8 || in_nbit == 16, BF_STATUS_UNSUPPORTED_DTYPE);
        BF_ASSERT(out_nbit == 32, BF_STATUS_UNSUPPORTED_DTYPE);
        bool sgn = (in->dtype & BF_DTYPE_TYPE_BITS) == BF_DTYPE_INT_TYPE;
        plan->int_in = true;
        plan->int_scale = 1.0f / (float)(1u << (in_nbit - (sgn ? 1 : 0)));
        plan->in_elem_nbyte = in_nbit / 8;
        f64 = false;
    } else {
        BF_ASSERT(in_nbit == out_nbit, BF_STATUS_UNSUPPORTED_DTYPE);
        BF_ASSERT(in_nbit == 32 || in_nbit == 64,
                  BF_STATUS_UNSUPPORTED_DTYPE);
        f64 = in_nbit == 64;
        plan->in_elem_nbyte = (in_nbit / 8) * (in_cplx ? 2 : 1);
    }
    plan->f64 = f64;
    plan->r2c = !in_cplx;
    plan->c2r = !out_cplx;
    BF_ASSERT(!(plan->r2c && plan->c2r), BF_STATUS_INVALID_ARGUMENT);
    BF_ASSERT(!(plan->int_in && plan->c2r), BF_STATUS_UNSUPPORTED_DTYPE);
    if (plan->r2c) plan->type = f64 ? HIPFFT_D2Z : HIPFFT_R2C;
    else if (plan->c2r) plan->type = f64 ? HIPFFT_Z2D : HIPFFT_C2R;
    else plan->type = f64 ? HIPFFT_Z2Z : HIPFFT_C2C;
    plan->do_shift = apply_fftshift;
    // fftshift is defined on the full (c2c) spectrum only; the reference
    // tests it only for c2c as well (test_fft.py run_test_c2c)
    BF_ASSERT(!apply_fftshift || (!plan->r2c && !plan->c2r),
              BF_STATUS_UNSUPPORTED);

    int nd = in->ndim;
    std::vector<int> ax(axes, axes + ndim);
    for (auto& a : ax) {
        if (a < 0) a += nd;
        BF_ASSERT(a >= 0 && a < nd, BF_STATUS_INVALID_ARGUMENT);
    }
    for (int i = 1; i < ndim; ++i)
        BF_ASSERT(ax[i] > ax[i - 1], BF_STATUS_UNSUPPORTED);
    int a_last = ax[ndim - 1];

    // logical transform lengths; r2c/c2r relate shapes on the last axis
    long n[3];
    for (int i = 0; i < ndim; ++i)
        n[i] = plan->c2r ? out->shape[ax[i]] : in->shape[ax[i]];
    if (plan->r2c) {
        BF_ASSERT(out->shape[a_last] == in->shape[a_last] / 2 + 1,
                  BF_STATUS_INVALID_SHAPE);
    } else if (plan->c2r) {
        BF_ASSERT(in->shape[a_last] == out->shape[a_last] / 2 + 1,
                  BF_STATUS_INVALID_SHAPE);
    }
    for (int d = 0; d < nd; ++d) {
        bool is_ax = false;
        for (int i = 0; i < ndim; ++i) is_ax |= (d == ax[i]);
        if (is_ax) {
            if (!plan->r2c && !plan->c2r)
                BF_ASSERT(out->shape[d] == in->shape[d],
                          BF_STATUS_INVALID_SHAPE);
        } else {
            BF_ASSERT(in->shape[d] == out->shape[d], BF_STATUS_INVALID_SHAPE);
        }
    }

    // element strides of the contiguous in/out arrays
    long istride_el[BF_MAX_DIMS], ostride_el[BF_MAX_DIMS];
    {
        long is = 1, os = 1;
        for (int d = nd - 1; d >= 0; --d) {
            istride_el[d] = is;
            ostride_el[d] = os;
            is *= in->shape[d];
            os *= out->shape[d];
        }
        plan->in_elems = (size_t)is;
        // spectrum side: out for forward/r2c, in for c2r
        plan->spec_elems = (size_t)(plan->c2r ? is : os);
    }

    // inner batch: dims after the last transform axis (idist = 1)
    long inner = 1;
    for (int d = a_last + 1; d < nd; ++d) inner *= in->shape[d];

    // inembed/onembed fold the gap dims: stride(ax_i) must equal
    // istride * prod_{j>i} embed[j]  (embed[0] is ignored by hipFFT)
    int nn[3], inembed[3], onembed[3];
    for (int i = 0; i < ndim; ++i) nn[i] = (int)n[i];
    for (int i = 1; i < ndim; ++i) {
        long ie = istride_el[ax[i - 1]] / istride_el[ax[i]];
        long oe = ostride_el[ax[i - 1]] / ostride_el[ax[i]];
        BF_ASSERT(istride_el[ax[i - 1]] % istride_el[ax[i]] == 0 &&
                  ostride_el[ax[i - 1]] % ostride_el[ax[i]] == 0,
                  BF_STATUS_UNSUPPORTED_STRIDE);
        inembed[i] = (int)ie;
        onembed[i] = (int)oe;
    }
    inembed[0] = nn[0];
    onembed[0] = nn[0];

    // loop dims = everything not a transform axis and not after a_last;
    // if inner == 1, promote ONE loop dim (the innermost) to the plan's
    // batch via idist = its stride.
    plan->loop_shape.clear();
    plan->loop_istride.clear();
    plan->loop_ostride.clear();
    plan->nloop = 1;
    std::vector<int> loop_dims;
    for (int d = 0; d < a_last; ++d) {
        bool is_ax = false;
        for (int i = 0; i < ndim; ++i) is_ax |= (d == ax[i]);
        if (!is_ax) loop_dims.push_back(d);
    }
    long batch, idist, odist;
    if (inner > 1 || loop_dims.empty()) {
        batch = inner > 0 ? inner : 1;
        idist = 1;
        odist = 1;
    } else {
        int bd = loop_dims.back();  // innermost loop dim: smallest stride
        loop_dims.pop_back();
        batch = in->shape[bd];
        idist = istride_el[bd];
        odist = ostride_el[bd];
    }
    for (int d : loop_dims) {
        plan->loop_shape.push_back(in->shape[d]);
        plan->loop_istride.push_back(istride_el[d]);
        plan->loop_ostride.push_back(ostride_el[d]);
        plan->nloop *= in->shape[d];
    }

    long istride = istride_el[a_last];
    long ostride = ostride_el[a_last];

    if (plan->have_plan) {
        hipfftDestroy(plan->plan);
        plan->have_plan = false;
    }
    BF_CHECK_HIPFFT(hipfftCreate(&plan->plan));
    plan->have_plan = true;
    size_t worksize = 0;
    BF_CHECK_HIPFFT(hipfftMakePlanMany(
        plan->plan, ndim, nn, inembed, (int)istride, idist, onembed,
        (int)ostride, odist, plan->type, (int)batch, &worksize));
    plan->rank = ndim;
    plan->ndim = nd;
    for (int d = 0; d < nd; ++d) plan->ishape[d] = in->shape[d];
    for (int i = 0; i < ndim; ++i) plan->axlist[i] = ax[i];

    // roll geometry over the FULL spectrum-side array; shift only on the
    // transform axes.  Forward: fftshift out (shift = n - n/2 so that
    // out[k] = fft[(k + ceil(n/2)) % n]); inverse: ifftshift in
    // (in_unshifted[k] = in[(k + n/2) % n]).
    const BFarray* spec = plan->c2r ? in : out;
    plan->roll.ndim = nd;
    for (int d = 0; d < nd; ++d) {
        plan->roll.shape[d] = spec->shape[d];
        plan->roll.shift[d] = 0;
    }

    // temp buffers
    if (plan->tmp_conv) { hipFree(plan->tmp_conv); plan->tmp_conv = nullptr; }
    if (plan->tmp_roll) { hipFree(plan->tmp_roll); plan->tmp_roll = nullptr; }
    if (plan->int_in) {
        plan->tmp_conv_size = plan->in_elems * sizeof(float);
        BF_CHECK_HIP(hipMalloc(&plan->tmp_conv, plan->tmp_conv_size));
    }
    if (plan->do_shift) {
        // spectrum side is always complex (c2c-only shift)
        plan->tmp_roll_size = plan->spec_elems * (size_t)(f64 ? 16 : 8);
        BF_CHECK_HIP(hipMalloc(&plan->tmp_roll, plan->tmp_roll_size));
    }
    if (tmp_storage_size) *tmp_storage_size = 0;  // plan owns its temps
    return BF_STATUS_SUCCESS;
}

BFstatus bfFftExecute(BFfft plan, BFarray const* in, BFarray const* out,
                      BFbool inverse, void* tmp_storage,
                      size_t tmp_storage_size) {
    using namespace bfamd;
    (void)tmp_storage;
    (void)tmp_storage_size;
    BF_ASSERT(plan && plan->have_plan, BF_STATUS_INVALID_HANDLE);
    BF_ASSERT(in && out, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(in->ndim == plan->ndim, BF_STATUS_INVALID_SHAPE);
    for (int d = 0; d < plan->ndim; ++d)
        BF_ASSERT(in->shape[d] == plan->ishape[d], BF_STATUS_INVALID_SHAPE);

    hipStream_t s = bfamd::thread_stream();
    BF_CHECK_HIPFFT(hipfftSetStream(plan->plan, s));

    int cbytes = plan->f64 ? 16 : 8;  // complex element bytes
    int rbytes = plan->f64 ? 8 : 4;

    // 1. integer conversion (whole array)
    const void* src = in->data;
    int src_elem_nbyte = plan->in_elem_nbyte;
    if (plan->int_in) {
        size_t nelem = plan->in_elems;
        if (plan->in_elem_nbyte == 1) {
            if (plan->int_scale == 1.0f / 256.0f)
                hipLaunchKernelGGL(convert_kernel<unsigned char>,
                                   dim3(grid_for(nelem)), dim3(256), 0, s,
                                   (const unsigned char*)in->data,
                                   (float*)plan->tmp_conv, plan->int_scale,
                                   nelem);
            else
                hipLaunchKernelGGL(convert_kernel<signed char>,
                                   dim3(grid_for(nelem)), dim3(256), 0, s,
                                   (const signed char*)in->data,
                                   (float*)plan->tmp_conv, plan->int_scale,
                                   nelem);
        } else {
            if (plan->int_scale == 1.0f / 65536.0f)
                hipLaunchKernelGGL(convert_kernel<unsigned short>,
                                   dim3(grid_for(nelem)), dim3(256), 0, s,
                                   (const unsigned short*)in->data,
                                   (float*)plan->tmp_conv, plan->int_scale,
                                   nelem);
            else
                hipLaunchKernelGGL(convert_kernel<short>,
                                   dim3(grid_for(nelem)), dim3(256), 0, s,
                                   (const short*)in->data,
                                   (float*)plan->tmp_conv, plan->int_scale,
                                   nelem);
        }
        BF_CHECK_HIP(hipGetLastError());
        src = plan->tmp_conv;
        src_elem_nbyte = 4;
    }

    // spectrum-side shift (c2c only): forward rolls the OUTPUT by
    // fftshift after the transform; inverse rolls the INPUT by ifftshift
    // before it.  Gather shifts: fftshift out[k] = z[(k + n - n/2) % n],
    // ifftshift y[k] = x[(k + n/2) % n].
    bool pre_roll = plan->do_shift && inverse;
    bool post_roll = plan->do_shift && !inverse;
    void* fft_out = out->data;

    if (pre_roll) {
        RollArgs ra = plan->roll;
        for (int i = 0; i < plan->rank; ++i) {
            int d = plan->axlist[i];
            ra.shift[d] = ra.shape[d] / 2;  // ifftshift gather
        }
        size_t n = plan->spec_elems;
        if (plan->f64)
            hipLaunchKernelGGL((roll_kernel<double2>), dim3(grid_for(n)),
                               dim3(256), 0, s, (const double2*)src,
                               (double2*)plan->tmp_roll, ra, n);
        else
            hipLaunchKernelGGL((roll_kernel<float2>), dim3(grid_for(n)),
                               dim3(256), 0, s, (const float2*)src,
                               (float2*)plan->tmp_roll, ra, n);
        BF_CHECK_HIP(hipGetLastError());
        src = plan->tmp_roll;
    }
    if (post_roll) fft_out = plan->tmp_roll;

    int dir = inverse ? HIPFFT_BACKWARD : HIPFFT_FORWARD;

    // iterate host loops
    long nloop = plan->nloop;
    std::vector<long> idx(plan->loop_shape.size(), 0);
    for (long l = 0; l < nloop; ++l) {
        long ioff = 0, ooff = 0;
        for (size_t d = 0; d < idx.size(); ++d) {
            ioff += idx[d] * plan->loop_istride[d];
            ooff += idx[d] * plan->loop_ostride[d];
        }
        char* ip = (char*)src +
                   ioff * (plan->r2c ? src_elem_nbyte : cbytes);
        char* op = (char*)fft_out +
                   ooff * (plan->c2r ? rbytes : cbytes);
        if (plan->r2c) {
            if (plan->f64)
                BF_CHECK_HIPFFT(hipfftExecD2Z(plan->plan, (double*)ip,
                                              (hipfftDoubleComplex*)op));
            else
                BF_CHECK_HIPFFT(hipfftExecR2C(plan->plan, (float*)ip,
                                              (hipfftComplex*)op));
        } else if (plan->c2r) {
            if (plan->f64)
                BF_CHECK_HIPFFT(hipfftExecZ2D(plan->plan,
                                              (hipfftDoubleComplex*)ip,
                                              (double*)op));
            else
                BF_CHECK_HIPFFT(hipfftExecC2R(plan->plan, (hipfftComplex*)ip,
                                              (float*)op));
        } else {
            if (plan->f64)
                BF_CHECK_HIPFFT(hipfftExecZ2Z(plan->plan,
                                              (hipfftDoubleComplex*)ip,
                                              (hipfftDoubleComplex*)op, dir));
            else
                BF_CHECK_HIPFFT(hipfftExecC2C(plan->plan, (hipfftComplex*)ip,
                                              (hipfftComplex*)op, dir));
        }
        // advance multi-index
        for (int d = (int)idx.size() - 1; d >= 0; --d) {
            if (++idx[d] < plan->loop_shape[d]) break;
            idx[d] = 0;
        }
    }

    if (post_roll) {
        RollArgs ra = plan->roll;
        for (int i = 0; i < plan->rank; ++i) {
            int d = plan->axlist[i];
            ra.shift[d] = ra.shape[d] - ra.shape[d] / 2;  // fftshift gather
        }
        size_t n = plan->spec_elems;
        if (plan->f64)
            hipLaunchKernelGGL((roll_kernel<double2>), dim3(grid_for(n)),
                               dim3(256), 0, s,
                               (const double2*)plan->tmp_roll,
                               (double2*)out->data, ra, n);
        else
            hipLaunchKernelGGL((roll_kernel<float2>), dim3(grid_for(n)),
                               dim3(256), 0, s,
                               (const float2*)plan->tmp_roll,
                               (float2*)out->data, ra, n);
        BF_CHECK_HIP(hipGetLastError());
    }
    return BF_STATUS_SUCCESS;
}

}  // extern "C"
