// bifrost_amd: status plumbing.  Exceptions never cross the C ABI: every
// entry point converts to BFstatus (reference contract).
#pragma once

#include <bifrost/common.h>

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <exception>
#include <new>
#include <stdexcept>

namespace bfamd {

// Throwing helper used inside implementation code.
struct StatusError : public std::exception {
    BFstatus status;
    explicit StatusError(BFstatus s) : status(s) {}
    const char* what() const noexcept override { return "bifrost status error"; }
};

bool debug_enabled();

}  // namespace bfamd

#define BF_ASSERT(cond, err)                                              \
    do {                                                                  \
        if (!(cond)) {                                                    \
            if (bfamd::debug_enabled()) {                                 \
                std::fprintf(stderr, "[bifrost_amd] %s:%d: assert failed:"\
                             " %s -> %d\n", __FILE__, __LINE__, #cond,    \
                             (int)(err));                                 \
            }                                                             \
            return (err);                                                 \
        }                                                                 \
    } while (0)

// Variant for use inside functions that throw instead of returning status.
#define BF_THROW_IF(cond, err)                                            \
    do {                                                                  \
        if (cond) throw bfamd::StatusError(err);                          \
    } while (0)

#define BF_CHECK(call)                                                    \
    do {                                                                  \
        BFstatus bf_st_ = (call);                                         \
        if (bf_st_ != BF_STATUS_SUCCESS) return bf_st_;                   \
    } while (0)

#define BF_CHECK_HIP(call)                                                \
    do {                                                                  \
        hipError_t hip_st_ = (call);                                      \
        if (hip_st_ != hipSuccess) {                                      \
            if (bfamd::debug_enabled()) {                                 \
                std::fprintf(stderr, "[bifrost_amd] %s:%d: HIP error: %s\n",\
                             __FILE__, __LINE__,                          \
                             hipGetErrorString(hip_st_));                 \
            }                                                             \
            return BF_STATUS_DEVICE_ERROR;                                \
        }                                                                 \
    } while (0)

#define BF_CHECK_HIP_THROW(call)                                          \
    do {                                                                  \
        hipError_t hip_st_ = (call);                                      \
        if (hip_st_ != hipSuccess) {                                      \
            if (bfamd::debug_enabled()) {                                 \
                std::fprintf(stderr, "[bifrost_amd] %s:%d: HIP error: %s\n",\
                             __FILE__, __LINE__,                          \
                             hipGetErrorString(hip_st_));                 \
            }                                                             \
            throw bfamd::StatusError(BF_STATUS_DEVICE_ERROR);             \
        }                                                                 \
    } while (0)

// Wrap a C-ABI body that may throw.  Variadic so brace-initializers with
// commas survive preprocessing.
#define BF_TRY_RETURN(...)                                                \
    try {                                                                 \
        __VA_ARGS__;                                                      \
        return BF_STATUS_SUCCESS;                                         \
    } catch (bfamd::StatusError & e) {                                    \
        return e.status;                                                  \
    } catch (std::bad_alloc&) {                                           \
        return BF_STATUS_MEM_ALLOC_FAILED;                                \
    } catch (std::exception&) {                                           \
        return BF_STATUS_INTERNAL_ERROR;                                  \
    }
