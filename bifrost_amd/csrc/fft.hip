// bifrost_amd: bfFft — FFTs over BFarray axes on hipFFT/rocFFT
// (SURVEY.md §8f row n2).  Behaviour contract: reference src/fft.cu
// semantics — unnormalized transforms (cuFFT convention; numpy's ifft
// normalization is NOT applied), c2c/r2c/c2r by dtype pairing, transforms
// over an arbitrary set of CONSECUTIVE axes of a contiguous array (outer
// batch dims looped, inner batch expressed through the plan's
// stride/dist embedding).  apply_fftshift is not supported this round
// (reference implements it with cuFFT load callbacks).

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

struct BFfft_impl {
    hipfftHandle plan = 0;
    bool have_plan = false;
    hipfftType type = HIPFFT_C2C;
    int rank = 0;
    long nloop = 1;            // outer batch iterations
    long iloop_stride = 0;     // elements between outer batches (in)
    long oloop_stride = 0;
    bool f64 = false;
    bool r2c = false, c2r = false;
    // recorded for shape verification at execute time
    int ndim = 0;
    long ishape[BF_MAX_DIMS] = {0};

    ~BFfft_impl() {
        if (have_plan) hipfftDestroy(plan);
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
    BF_ASSERT(!apply_fftshift, BF_STATUS_UNSUPPORTED);
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
    BF_ASSERT(in_nbit == out_nbit, BF_STATUS_UNSUPPORTED_DTYPE);
    BF_ASSERT(in_nbit == 32 || in_nbit == 64, BF_STATUS_UNSUPPORTED_DTYPE);
    bool f64 = in_nbit == 64;
    plan->f64 = f64;
    plan->r2c = !in_cplx;
    plan->c2r = !out_cplx;
    if (plan->r2c) plan->type = f64 ? HIPFFT_D2Z : HIPFFT_R2C;
    else if (plan->c2r) plan->type = f64 ? HIPFFT_Z2D : HIPFFT_C2R;
    else plan->type = f64 ? HIPFFT_Z2Z : HIPFFT_C2C;

    int nd = in->ndim;
    // normalize + sort axes (must be consecutive, ascending)
    std::vector<int> ax(axes, axes + ndim);
    for (auto& a : ax) {
        if (a < 0) a += nd;
        BF_ASSERT(a >= 0 && a < nd, BF_STATUS_INVALID_ARGUMENT);
    }
    for (int i = 1; i < ndim; ++i)
        BF_ASSERT(ax[i] == ax[i - 1] + 1, BF_STATUS_UNSUPPORTED);
    int a0 = ax[0], a1 = ax[ndim - 1];

    // logical transform lengths (from the input for c2c/r2c, from the
    // output for c2r, where the real length defines n)
    long n[3];
    for (int i = 0; i < ndim; ++i) {
        n[i] = plan->c2r ? out->shape[ax[i]] : in->shape[ax[i]];
    }
    // r2c/c2r: the transformed fastest axis halves (+1) on the complex side
    if (plan->r2c) {
        BF_ASSERT(out->shape[a1] == in->shape[a1] / 2 + 1,
                  BF_STATUS_INVALID_SHAPE);
    } else if (plan->c2r) {
        BF_ASSERT(in->shape[a1] == out->shape[a1] / 2 + 1,
                  BF_STATUS_INVALID_SHAPE);
    } else {
        for (int i = 0; i < ndim; ++i)
            BF_ASSERT(out->shape[ax[i]] == in->shape[ax[i]],
                      BF_STATUS_INVALID_SHAPE);
    }

    // batch structure: inner = product of dims AFTER a1 (handled via
    // stride/dist embedding), outer = product of dims BEFORE a0 (looped).
    long inner = 1, outer = 1;
    for (int d = a1 + 1; d < nd; ++d) {
        BF_ASSERT(in->shape[d] == out->shape[d], BF_STATUS_INVALID_SHAPE);
        inner *= in->shape[d];
    }
    for (int d = 0; d < a0; ++d) {
        BF_ASSERT(in->shape[d] == out->shape[d], BF_STATUS_INVALID_SHAPE);
        outer *= in->shape[d];
    }
    int nn[3];
    for (int i = 0; i < ndim; ++i) nn[i] = (int)n[i];

    // Embeddings: transforms cover dims [a0, a1]; elements of the
    // transform block are strided by `inner`; batches: inner batch count =
    // inner with dist 1; outer batches looped at execute.
    long in_block = 1, out_block = 1;  // elements per transform block incl.
    for (int i = 0; i < ndim; ++i) {
        in_block *= in->shape[ax[i]];
        out_block *= out->shape[ax[i]];
    }
    int inembed[3], onembed[3];
    for (int i = 0; i < ndim; ++i) {
        inembed[i] = (int)in->shape[ax[i]];
        onembed[i] = (int)out->shape[ax[i]];
    }

    if (plan->have_plan) {
        hipfftDestroy(plan->plan);
        plan->have_plan = false;
    }
    BF_CHECK_HIPFFT(hipfftCreate(&plan->plan));
    plan->have_plan = true;
    size_t worksize = 0;
    BF_CHECK_HIPFFT(hipfftMakePlanMany(
        plan->plan, ndim, nn, inembed, (int)inner, 1, onembed, (int)inner, 1,
        plan->type, (int)inner, &worksize));
    plan->rank = ndim;
    plan->nloop = outer;
    plan->iloop_stride = in_block * inner;
    plan->oloop_stride = out_block * inner;
    plan->ndim = nd;
    for (int d = 0; d < nd; ++d) plan->ishape[d] = in->shape[d];
    if (tmp_storage_size) *tmp_storage_size = 0;  // hipfft manages its own
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

    BF_CHECK_HIPFFT(hipfftSetStream(plan->plan, bfamd::thread_stream()));
    int dir = inverse ? HIPFFT_BACKWARD : HIPFFT_FORWARD;
    int ies = plan->f64 ? 16 : 8;   // complex element bytes
    int res = plan->f64 ? 8 : 4;    // real element bytes
    for (long l = 0; l < plan->nloop; ++l) {
        char* ip = (char*)in->data +
                   l * plan->iloop_stride * (plan->r2c ? res : ies);
        char* op = (char*)out->data +
                   l * plan->oloop_stride * (plan->c2r ? res : ies);
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
    }
    return BF_STATUS_SUCCESS;
}

}  // extern "C"
