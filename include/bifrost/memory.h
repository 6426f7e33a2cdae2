/* bifrost_amd: space-aware memory management over HIP.
 *
 * BFspace enum values and the space strings (incl. the literal "cuda",
 * which maps to hipMalloc device memory here) reproduce reference
 * src/bifrost/memory.h:44-50 / memory.cpp:94-106 so existing pipelines
 * with space='cuda' land on HIP unchanged.
 */
#ifndef BFAMD_MEMORY_H_
#define BFAMD_MEMORY_H_

#include <bifrost/common.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum BFspace_ {
    BF_SPACE_AUTO         = 0,
    BF_SPACE_SYSTEM       = 1,  /* aligned_alloc            */
    BF_SPACE_CUDA         = 2,  /* hipMalloc                */
    BF_SPACE_CUDA_HOST    = 3,  /* hipHostMalloc            */
    BF_SPACE_CUDA_MANAGED = 4   /* hipMallocManaged         */
} BFspace;

BFstatus bfMalloc(void** ptr, BFsize size, BFspace space);
BFstatus bfFree(void* ptr, BFspace space);
BFstatus bfGetSpace(const void* ptr, BFspace* space);
const char* bfGetSpaceString(BFspace space);

/* Sync wrt host, async wrt device (enqueued on the calling thread's stream). */
BFstatus bfMemcpy(void* dst, BFspace dst_space,
                  const void* src, BFspace src_space, BFsize count);
BFstatus bfMemcpy2D(void* dst, BFsize dst_stride, BFspace dst_space,
                    const void* src, BFsize src_stride, BFspace src_space,
                    BFsize width, BFsize height);
BFstatus bfMemset(void* ptr, BFspace space, int value, BFsize count);
BFstatus bfMemset2D(void* ptr, BFsize stride, BFspace space,
                    int value, BFsize width, BFsize height);
BFsize bfGetAlignment(void);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_MEMORY_H_ */
