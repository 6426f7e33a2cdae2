/* bifrost_amd: the linalg hot path (per-channel cross-correlation "cherk"
 * and small-N beamforming cgemm on hand-written CDNA4 HIP kernels).
 * ABI identical to reference src/bifrost/linalg.h:43-57.
 */
#ifndef BFAMD_LINALG_H_
#define BFAMD_LINALG_H_

#include <bifrost/common.h>
#include <bifrost/array.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct BFlinalg_impl* BFlinalg;

BFstatus bfLinAlgCreate(BFlinalg* handle_ptr);
BFstatus bfLinAlgDestroy(BFlinalg handle);

/* c = alpha*a.b + beta*c; or alpha*a.a^H (b NULL) / alpha*b^H.b (a NULL).
 * Row-major semantics as numpy.matmul over the last two dims; leading dims
 * batch.  For the a^H.a / a.a^H forms only the LOWER triangle of c is
 * written (matrix_fill_mode 'lower').
 */
BFstatus bfLinAlgMatMul(BFlinalg       handle,
                        double         alpha,
                        BFarray const* a,   /* [...,i,j] */
                        BFarray const* b,   /* [...,j,k] */
                        double         beta,
                        BFarray const* c);  /* [...,i,k] */

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_LINALG_H_ */
