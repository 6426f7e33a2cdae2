/* bifrost_amd: bfMap — JIT'd user functions over ndarrays (hipRTC).
 * ABI identical to reference src/bifrost/map.h:40-63. */
#ifndef BFAMD_MAP_H_
#define BFAMD_MAP_H_

#include <bifrost/common.h>
#include <bifrost/array.h>

#ifdef __cplusplus
extern "C" {
#endif

BFstatus bfMap(int                  ndim,
               long const*          shape,
               char const*const*    axis_names,
               int                  narg,
               BFarray const*const* args,
               char const*const*    arg_names,
               char const*          func_name,
               char const*          func,
               char const*          extra_code,
               int const*           block_shape,  /* length 2 or NULL */
               int const*           block_axes);  /* length 2 or NULL */
BFstatus bfMapClearCache(void);

#define BF_MAP_KERNEL_CACHE_SIZE 128

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_MAP_H_ */
