/* bifrost_amd: MI355X-native libbifrost — common definitions.
 *
 * Drop-in replacement for the reference C ABI: the typedefs and the
 * BFstatus enum values reproduce reference src/bifrost/common.h:41-76
 * exactly (they are the wire contract consumed by ctypes bindings).
 */
#ifndef BFAMD_COMMON_H_
#define BFAMD_COMMON_H_

#ifdef __cplusplus
extern "C" {
#endif

typedef int                BFbool;
typedef float              BFcomplex[2];
typedef float              BFreal;
typedef unsigned long      BFsize;
typedef unsigned long long BFoffset;
typedef   signed long long BFdelta;

typedef enum BFstatus_ {
    BF_STATUS_SUCCESS              = 0,
    BF_STATUS_END_OF_DATA          = 1,
    BF_STATUS_WOULD_BLOCK          = 2,
    BF_STATUS_INVALID_POINTER      = 8,
    BF_STATUS_INVALID_HANDLE       = 9,
    BF_STATUS_INVALID_ARGUMENT     = 10,
    BF_STATUS_INVALID_STATE        = 11,
    BF_STATUS_INVALID_SPACE        = 12,
    BF_STATUS_INVALID_SHAPE        = 13,
    BF_STATUS_INVALID_STRIDE       = 14,
    BF_STATUS_INVALID_DTYPE        = 15,
    BF_STATUS_MEM_ALLOC_FAILED     = 32,
    BF_STATUS_MEM_OP_FAILED        = 33,
    BF_STATUS_UNSUPPORTED          = 48,
    BF_STATUS_UNSUPPORTED_SPACE    = 49,
    BF_STATUS_UNSUPPORTED_SHAPE    = 50,
    BF_STATUS_UNSUPPORTED_STRIDE   = 51,
    BF_STATUS_UNSUPPORTED_DTYPE    = 52,
    BF_STATUS_FAILED_TO_CONVERGE   = 64,
    BF_STATUS_INSUFFICIENT_STORAGE = 65,
    BF_STATUS_DEVICE_ERROR         = 66,
    BF_STATUS_INTERNAL_ERROR       = 99
} BFstatus;

const char* bfGetStatusString(BFstatus status);
BFbool      bfGetDebugEnabled(void);
BFstatus    bfSetDebugEnabled(BFbool enabled);
/* Kept truthy on ROCm: "cuda" is the device-space alias (see memory.h). */
BFbool      bfGetCudaEnabled(void);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_COMMON_H_ */
