// bifrost_amd: space-aware allocation and copies on HIP.
// ABI: reference src/bifrost/memory.h:52-85; the "cuda" space names map to
// HIP allocations (memory.h comment).  Copies enqueue async on the calling
// thread's stream and are synchronized before return for host-visible
// destinations only when required by the HIP API (hipMemcpyAsync on
// pageable host memory is effectively synchronous).

#include <bifrost/memory.h>

#include <cstdlib>
#include <cstring>

#include "dtype.hpp"
#include "hipctx.hpp"
#include "status.hpp"

namespace {
constexpr BFsize kAlignment = 4096;  // page-aligned; >= any SIMD need

hipMemcpyKind copy_kind(BFspace dst, BFspace src) {
    bool d_dev = bfamd::space_on_device(dst);
    bool s_dev = bfamd::space_on_device(src);
    if (d_dev && s_dev) return hipMemcpyDeviceToDevice;
    if (d_dev) return hipMemcpyHostToDevice;
    if (s_dev) return hipMemcpyDeviceToHost;
    return hipMemcpyHostToHost;
}

bool pure_host(BFspace dst, BFspace src) {
    return dst == BF_SPACE_SYSTEM && src == BF_SPACE_SYSTEM;
}
}  // namespace

extern "C" {

BFstatus bfMalloc(void** ptr, BFsize size, BFspace space) {
    BF_ASSERT(ptr, BF_STATUS_INVALID_POINTER);
    switch (space) {
        case BF_SPACE_AUTO:
        case BF_SPACE_SYSTEM: {
            BFsize padded = (size + kAlignment - 1) / kAlignment * kAlignment;
            void* p = std::aligned_alloc(kAlignment, padded ? padded : kAlignment);
            BF_ASSERT(p, BF_STATUS_MEM_ALLOC_FAILED);
            *ptr = p;
            return BF_STATUS_SUCCESS;
        }
        case BF_SPACE_CUDA:
            BF_CHECK_HIP(hipMalloc(ptr, size ? size : 1));
            return BF_STATUS_SUCCESS;
        case BF_SPACE_CUDA_HOST:
            BF_CHECK_HIP(hipHostMalloc(ptr, size ? size : 1, hipHostMallocDefault));
            return BF_STATUS_SUCCESS;
        case BF_SPACE_CUDA_MANAGED:
            BF_CHECK_HIP(hipMallocManaged(ptr, size ? size : 1));
            return BF_STATUS_SUCCESS;
        default:
            return BF_STATUS_INVALID_SPACE;
    }
}

BFstatus bfFree(void* ptr, BFspace space) {
    BF_ASSERT(ptr, BF_STATUS_INVALID_POINTER);
    switch (space) {
        case BF_SPACE_AUTO:
        case BF_SPACE_SYSTEM: std::free(ptr); return BF_STATUS_SUCCESS;
        case BF_SPACE_CUDA:
            BF_CHECK_HIP(hipFree(ptr)); return BF_STATUS_SUCCESS;
        case BF_SPACE_CUDA_HOST:
            BF_CHECK_HIP(hipHostFree(ptr)); return BF_STATUS_SUCCESS;
        case BF_SPACE_CUDA_MANAGED:
            BF_CHECK_HIP(hipFree(ptr)); return BF_STATUS_SUCCESS;
        default:
            return BF_STATUS_INVALID_SPACE;
    }
}

BFstatus bfGetSpace(const void* ptr, BFspace* space) {
    BF_ASSERT(ptr && space, BF_STATUS_INVALID_POINTER);
    hipPointerAttribute_t attr;
    hipError_t err = hipPointerGetAttributes(&attr, ptr);
    if (err != hipSuccess) {
        (void)hipGetLastError();  // clear
        *space = BF_SPACE_SYSTEM;
        return BF_STATUS_SUCCESS;
    }
    switch (attr.type) {
        case hipMemoryTypeDevice:  *space = BF_SPACE_CUDA; break;
        case hipMemoryTypeHost:    *space = BF_SPACE_CUDA_HOST; break;
        case hipMemoryTypeManaged: *space = BF_SPACE_CUDA_MANAGED; break;
        case hipMemoryTypeUnified: *space = BF_SPACE_CUDA_MANAGED; break;
        default:                   *space = BF_SPACE_SYSTEM; break;
    }
    return BF_STATUS_SUCCESS;
}

const char* bfGetSpaceString(BFspace space) {
    // String values are ABI (reference memory.cpp:94-106): user pipelines
    // pass space='cuda' and it must land on HIP device memory here.
    switch (space) {
        case BF_SPACE_AUTO:         return "auto";
        case BF_SPACE_SYSTEM:       return "system";
        case BF_SPACE_CUDA:         return "cuda";
        case BF_SPACE_CUDA_HOST:    return "cuda_host";
        case BF_SPACE_CUDA_MANAGED: return "cuda_managed";
        default:                    return "unknown";
    }
}

BFstatus bfMemcpy(void* dst, BFspace dst_space,
                  const void* src, BFspace src_space, BFsize count) {
    BF_ASSERT(dst && src, BF_STATUS_INVALID_POINTER);
    if (count == 0) return BF_STATUS_SUCCESS;
    if (pure_host(dst_space, src_space)) {
        std::memcpy(dst, src, count);
        return BF_STATUS_SUCCESS;
    }
    BF_CHECK_HIP(hipMemcpyAsync(dst, src, count,
                                copy_kind(dst_space, src_space),
                                bfamd::thread_stream()));
    if (!bfamd::space_on_device(dst_space)) {
        // sync wrt host when the destination is host-visible
        BF_CHECK_HIP(hipStreamSynchronize(bfamd::thread_stream()));
    }
    return BF_STATUS_SUCCESS;
}

BFstatus bfMemcpy2D(void* dst, BFsize dst_stride, BFspace dst_space,
                    const void* src, BFsize src_stride, BFspace src_space,
                    BFsize width, BFsize height) {
    BF_ASSERT(dst && src, BF_STATUS_INVALID_POINTER);
    if (width == 0 || height == 0) return BF_STATUS_SUCCESS;
    if (pure_host(dst_space, src_space)) {
        for (BFsize r = 0; r < height; ++r) {
            std::memcpy((char*)dst + r * dst_stride,
                        (const char*)src + r * src_stride, width);
        }
        return BF_STATUS_SUCCESS;
    }
    BF_CHECK_HIP(hipMemcpy2DAsync(dst, dst_stride, src, src_stride,
                                  width, height,
                                  copy_kind(dst_space, src_space),
                                  bfamd::thread_stream()));
    if (!bfamd::space_on_device(dst_space)) {
        BF_CHECK_HIP(hipStreamSynchronize(bfamd::thread_stream()));
    }
    return BF_STATUS_SUCCESS;
}

BFstatus bfMemset(void* ptr, BFspace space, int value, BFsize count) {
    BF_ASSERT(ptr, BF_STATUS_INVALID_POINTER);
    if (count == 0) return BF_STATUS_SUCCESS;
    if (space == BF_SPACE_SYSTEM || space == BF_SPACE_AUTO) {
        std::memset(ptr, value, count);
        return BF_STATUS_SUCCESS;
    }
    BF_CHECK_HIP(hipMemsetAsync(ptr, value, count, bfamd::thread_stream()));
    return BF_STATUS_SUCCESS;
}

BFstatus bfMemset2D(void* ptr, BFsize stride, BFspace space,
                    int value, BFsize width, BFsize height) {
    BF_ASSERT(ptr, BF_STATUS_INVALID_POINTER);
    if (width == 0 || height == 0) return BF_STATUS_SUCCESS;
    if (space == BF_SPACE_SYSTEM || space == BF_SPACE_AUTO) {
        for (BFsize r = 0; r < height; ++r)
            std::memset((char*)ptr + r * stride, value, width);
        return BF_STATUS_SUCCESS;
    }
    BF_CHECK_HIP(hipMemset2DAsync(ptr, stride, value, width, height,
                                  bfamd::thread_stream()));
    return BF_STATUS_SUCCESS;
}

BFsize bfGetAlignment(void) { return kAlignment; }

}  // extern "C"
