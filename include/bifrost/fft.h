/* bifrost_amd: bfFft — FFTs over array axes (hipFFT/rocFFT backend).
 * ABI identical to reference src/bifrost/fft.h:40-62; transforms are
 * unnormalized (cuFFT convention). */
#ifndef BFAMD_FFT_H_
#define BFAMD_FFT_H_

#include <bifrost/common.h>
#include <bifrost/array.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct BFfft_impl* BFfft;

BFstatus bfFftCreate(BFfft* plan_ptr);
BFstatus bfFftInit(BFfft          plan,
                   BFarray const* iarray,
                   BFarray const* oarray,
                   int            ndim,
                   int     const* axes,
                   BFbool         apply_fftshift,
                   size_t*        tmp_storage_size);
/* in,out = complex,complex => [i]fft; real,complex => rfft;
 * complex,real => irfft */
BFstatus bfFftExecute(BFfft          plan,
                      BFarray const* iarray,
                      BFarray const* oarray,
                      BFbool         inverse,
                      void*          tmp_storage,
                      size_t         tmp_storage_size);
BFstatus bfFftDestroy(BFfft plan);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_FFT_H_ */
