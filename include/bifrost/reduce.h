/* bifrost_amd: axis reductions (scrunching) on device arrays.
 * ABI identical to reference src/bifrost/reduce.h:44-57 (enum values and
 * the bfReduce signature).  The reduced axis is inferred: the one dim
 * where out->shape[d] < in->shape[d] (must divide). */
#ifndef BFAMD_REDUCE_H_
#define BFAMD_REDUCE_H_

#include <bifrost/common.h>
#include <bifrost/array.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum BFreduce_op_ {
    BF_REDUCE_SUM,          /* sum(x) */
    BF_REDUCE_MEAN,         /* sum(x) / n */
    BF_REDUCE_MIN,          /* min(x) */
    BF_REDUCE_MAX,          /* max(x) */
    BF_REDUCE_STDERR,       /* sum(x) / sqrt(n) */
    BF_REDUCE_POWER_SUM,    /* sum(|x|^2) */
    BF_REDUCE_POWER_MEAN,   /* sum(|x|^2) / n */
    BF_REDUCE_POWER_MIN,    /* min(|x|^2) */
    BF_REDUCE_POWER_MAX,    /* max(|x|^2) */
    BF_REDUCE_POWER_STDERR  /* sum(|x|^2) / sqrt(n) */
} BFreduce_op;

BFstatus bfReduce(BFarray const* in, BFarray const* out, BFreduce_op op);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_REDUCE_H_ */
