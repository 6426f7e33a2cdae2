/* bifrost_amd: sub-byte sample unpacking (ci4 -> ci8/cf32 etc).
 * ABI identical to reference src/bifrost/unpack.h:37-40. */
#ifndef BFAMD_UNPACK_H_
#define BFAMD_UNPACK_H_

#include <bifrost/common.h>
#include <bifrost/array.h>

#ifdef __cplusplus
extern "C" {
#endif

BFstatus bfUnpack(BFarray const* in,
                  BFarray const* out,
                  BFbool         align_msb);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_UNPACK_H_ */
