/* bifrost_amd: CPU core affinity helpers.
 * ABI identical to reference src/bifrost/affinity.h:37-44. */
#ifndef BFAMD_AFFINITY_H_
#define BFAMD_AFFINITY_H_

#include <bifrost/common.h>

#ifdef __cplusplus
extern "C" {
#endif

/* core = -1 unbinds */
BFstatus bfAffinitySetCore(int core);
BFstatus bfAffinityGetCore(int* core);
BFstatus bfAffinitySetOpenMPCores(BFsize nthread, const int* thread_cores);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_AFFINITY_H_ */
