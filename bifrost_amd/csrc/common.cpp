// bifrost_amd: bfGetStatusString / debug flags / stream+device control.
// ABI: reference src/bifrost/common.h:79-84 and src/bifrost/cuda.h:36-46.

#include <bifrost/common.h>
#include <bifrost/cuda.h>

#include <atomic>
#include <cstdlib>
#include <cstring>

#include "hipctx.hpp"
#include "status.hpp"

namespace bfamd {

static std::atomic<int> g_debug{[] {
    const char* e = std::getenv("BIFROST_DEBUG");
    return (e && std::atoi(e)) ? 1 : 0;
}()};

bool debug_enabled() { return g_debug.load(std::memory_order_relaxed); }

hipStream_t& thread_stream() {
    static thread_local hipStream_t s = 0;
    return s;
}

// CPU-only hosts (no visible GPU) still run the system-space paths; stream
// sync becomes a no-op there instead of a device error.
bool hip_available() {
    static int count = [] {
        int c = 0;
        if (hipGetDeviceCount(&c) != hipSuccess) {
            (void)hipGetLastError();
            return 0;
        }
        return c;
    }();
    return count > 0;
}

}  // namespace bfamd

extern "C" {

const char* bfGetStatusString(BFstatus status) {
    switch (status) {
        case BF_STATUS_SUCCESS:              return "BF_STATUS_SUCCESS";
        case BF_STATUS_END_OF_DATA:          return "BF_STATUS_END_OF_DATA";
        case BF_STATUS_WOULD_BLOCK:          return "BF_STATUS_WOULD_BLOCK";
        case BF_STATUS_INVALID_POINTER:      return "BF_STATUS_INVALID_POINTER";
        case BF_STATUS_INVALID_HANDLE:       return "BF_STATUS_INVALID_HANDLE";
        case BF_STATUS_INVALID_ARGUMENT:     return "BF_STATUS_INVALID_ARGUMENT";
        case BF_STATUS_INVALID_STATE:        return "BF_STATUS_INVALID_STATE";
        case BF_STATUS_INVALID_SPACE:        return "BF_STATUS_INVALID_SPACE";
        case BF_STATUS_INVALID_SHAPE:        return "BF_STATUS_INVALID_SHAPE";
        case BF_STATUS_INVALID_STRIDE:       return "BF_STATUS_INVALID_STRIDE";
        case BF_STATUS_INVALID_DTYPE:        return "BF_STATUS_INVALID_DTYPE";
        case BF_STATUS_MEM_ALLOC_FAILED:     return "BF_STATUS_MEM_ALLOC_FAILED";
        case BF_STATUS_MEM_OP_FAILED:        return "BF_STATUS_MEM_OP_FAILED";
        case BF_STATUS_UNSUPPORTED:          return "BF_STATUS_UNSUPPORTED";
        case BF_STATUS_UNSUPPORTED_SPACE:    return "BF_STATUS_UNSUPPORTED_SPACE";
        case BF_STATUS_UNSUPPORTED_SHAPE:    return "BF_STATUS_UNSUPPORTED_SHAPE";
        case BF_STATUS_UNSUPPORTED_STRIDE:   return "BF_STATUS_UNSUPPORTED_STRIDE";
        case BF_STATUS_UNSUPPORTED_DTYPE:    return "BF_STATUS_UNSUPPORTED_DTYPE";
        case BF_STATUS_FAILED_TO_CONVERGE:   return "BF_STATUS_FAILED_TO_CONVERGE";
        case BF_STATUS_INSUFFICIENT_STORAGE: return "BF_STATUS_INSUFFICIENT_STORAGE";
        case BF_STATUS_DEVICE_ERROR:         return "BF_STATUS_DEVICE_ERROR";
        case BF_STATUS_INTERNAL_ERROR:       return "BF_STATUS_INTERNAL_ERROR";
        default:                             return "Unknown BFstatus";
    }
}

BFbool bfGetDebugEnabled(void) { return bfamd::debug_enabled(); }

BFstatus bfSetDebugEnabled(BFbool enabled) {
    bfamd::g_debug.store(enabled ? 1 : 0, std::memory_order_relaxed);
    return BF_STATUS_SUCCESS;
}

// "cuda" support means HIP device support on this backend.
BFbool bfGetCudaEnabled(void) { return 1; }

BFstatus bfStreamGet(void* stream) {
    BF_ASSERT(stream, BF_STATUS_INVALID_POINTER);
    *(hipStream_t*)stream = bfamd::thread_stream();
    return BF_STATUS_SUCCESS;
}

BFstatus bfStreamSet(void const* stream) {
    BF_ASSERT(stream, BF_STATUS_INVALID_POINTER);
    bfamd::thread_stream() = *(hipStream_t const*)stream;
    return BF_STATUS_SUCCESS;
}

BFstatus bfStreamSynchronize(void) {
    if (!bfamd::hip_available()) return BF_STATUS_SUCCESS;
    BF_CHECK_HIP(hipStreamSynchronize(bfamd::thread_stream()));
    return BF_STATUS_SUCCESS;
}

BFstatus bfDeviceGet(int* device) {
    BF_ASSERT(device, BF_STATUS_INVALID_POINTER);
    BF_CHECK_HIP(hipGetDevice(device));
    return BF_STATUS_SUCCESS;
}

BFstatus bfDeviceSet(int device) {
    BF_CHECK_HIP(hipSetDevice(device));
    return BF_STATUS_SUCCESS;
}

BFstatus bfDeviceSetById(const char* pci_bus_id) {
    BF_ASSERT(pci_bus_id, BF_STATUS_INVALID_POINTER);
    int device = -1;
    BF_CHECK_HIP(hipDeviceGetByPCIBusId(&device, pci_bus_id));
    BF_CHECK_HIP(hipSetDevice(device));
    return BF_STATUS_SUCCESS;
}

BFstatus bfDevicesSetNoSpinCPU(void) {
    int count = 0;
    BF_CHECK_HIP(hipGetDeviceCount(&count));
    int prev = 0;
    BF_CHECK_HIP(hipGetDevice(&prev));
    for (int d = 0; d < count; ++d) {
        BF_CHECK_HIP(hipSetDevice(d));
        BF_CHECK_HIP(hipSetDeviceFlags(hipDeviceScheduleBlockingSync));
    }
    BF_CHECK_HIP(hipSetDevice(prev));
    return BF_STATUS_SUCCESS;
}

}  // extern "C"
