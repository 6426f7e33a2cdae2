/* bifrost_amd: arbitrary-axes array transpose (HIP, LDS-tiled).
 * ABI identical to reference src/bifrost/transpose.h:37-41. */
#ifndef BFAMD_TRANSPOSE_H_
#define BFAMD_TRANSPOSE_H_

#include <bifrost/common.h>
#include <bifrost/array.h>

#ifdef __cplusplus
extern "C" {
#endif

BFstatus bfTranspose(BFarray const* in,
                     BFarray const* out,
                     int     const* axes);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_TRANSPOSE_H_ */
