/* bifrost_amd: float -> (complex) integer quantization with sub-byte packing.
 * ABI identical to reference src/bifrost/quantize.h:37-40. */
#ifndef BFAMD_QUANTIZE_H_
#define BFAMD_QUANTIZE_H_

#include <bifrost/common.h>
#include <bifrost/array.h>

#ifdef __cplusplus
extern "C" {
#endif

BFstatus bfQuantize(BFarray const* in,
                    BFarray const* out,
                    double         scale);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_QUANTIZE_H_ */
