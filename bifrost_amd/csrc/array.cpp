// bifrost_amd: BFarray allocation/copy/memset.
// ABI: reference src/bifrost/array.h:134-147 (bfArrayMalloc fills data +
// strides for a caller-specified space/dtype/shape; bfArrayCopy is a
// space-aware strided copy).

#include <bifrost/array.h>
#include <bifrost/memory.h>

#include "dtype.hpp"
#include "status.hpp"

using namespace bfamd;

extern "C" {

BFstatus bfArrayMalloc(BFarray* array) {
    BF_ASSERT(array, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(array->ndim >= 1 && array->ndim <= BF_MAX_DIMS,
              BF_STATUS_INVALID_SHAPE);
    int nbyte = dtype_nbyte(array->dtype);
    BF_ASSERT(nbyte > 0, BF_STATUS_UNSUPPORTED_DTYPE);
    long stride = nbyte;
    for (int d = array->ndim - 1; d >= 0; --d) {
        array->strides[d] = stride;
        stride *= array->shape[d];
    }
    return bfMalloc(&array->data, (BFsize)stride, array->space);
}

BFstatus bfArrayFree(const BFarray* array) {
    BF_ASSERT(array, BF_STATUS_INVALID_POINTER);
    return bfFree(array->data, array->space);
}

BFstatus bfArrayCopy(const BFarray* dst, const BFarray* src) {
    BF_ASSERT(dst && src, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(dst->data && src->data, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(!dst->immutable, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(shapes_equal(dst, src), BF_STATUS_INVALID_SHAPE);
    BF_ASSERT(dtype_nbit(dst->dtype) == dtype_nbit(src->dtype),
              BF_STATUS_UNSUPPORTED_DTYPE);

    if (is_contiguous(src) && is_contiguous(dst)) {
        return bfMemcpy(dst->data, dst->space, src->data, src->space,
                        capacity_bytes(src));
    }
    // Strided: fold the largest contiguous suffix of dims (identical in src
    // and dst) into one row, then 2D-copy over the remaining outer dims.
    int nbyte = dtype_nbyte(src->dtype);
    BF_ASSERT(nbyte > 0, BF_STATUS_UNSUPPORTED_DTYPE);
    long row_bytes = nbyte;
    int d = src->ndim;
    while (d > 0) {
        int i = d - 1;
        bool contig = src->strides[i] == dst->strides[i] &&
                      src->strides[i] == row_bytes;
        if (!contig) break;
        row_bytes *= src->shape[i];
        --d;
    }
    // dims [0, d) remain strided; dims [d, ndim) folded into row_bytes
    if (d == 0) {
        return bfMemcpy(dst->data, dst->space, src->data, src->space,
                        (BFsize)row_bytes);
    }
    --d;  // innermost remaining strided dim becomes the 2D height dim
    // Count rows over remaining dims; require them to be expressible as a
    // single height x pitch (one varying dim or contiguous combination).
    if (d == 0) {
        return bfMemcpy2D(dst->data, dst->strides[0], dst->space,
                          src->data, src->strides[0], src->space,
                          (BFsize)row_bytes, (BFsize)src->shape[0]);
    }
    // General ND fallback: iterate outer dims on the host.
    long counters[BF_MAX_DIMS] = {0};
    const char* sp = (const char*)src->data;
    char* dp = (char*)dst->data;
    for (;;) {
        long soff = 0, doff = 0;
        for (int i = 0; i <= d - 1; ++i) {
            soff += counters[i] * src->strides[i];
            doff += counters[i] * dst->strides[i];
        }
        BF_CHECK(bfMemcpy2D(dp + doff, dst->strides[d], dst->space,
                            sp + soff, src->strides[d], src->space,
                            (BFsize)row_bytes, (BFsize)src->shape[d]));
        int i = d - 1;
        for (; i >= 0; --i) {
            if (++counters[i] < src->shape[i]) break;
            counters[i] = 0;
        }
        if (i < 0) break;
    }
    return BF_STATUS_SUCCESS;
}

BFstatus bfArrayMemset(const BFarray* array, int value) {
    BF_ASSERT(array && array->data, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(!array->immutable, BF_STATUS_INVALID_POINTER);
    if (is_contiguous(array)) {
        return bfMemset(array->data, array->space, value,
                        capacity_bytes(array));
    }
    // Strided memset via 2D over the innermost contiguous row.
    int nbyte = dtype_nbyte(array->dtype);
    long row_bytes = array->shape[array->ndim - 1] * (long)nbyte;
    BF_ASSERT(array->ndim >= 2, BF_STATUS_UNSUPPORTED_STRIDE);
    BF_ASSERT(array->strides[array->ndim - 1] == nbyte,
              BF_STATUS_UNSUPPORTED_STRIDE);
    long nrow = 1;
    for (int i = 0; i < array->ndim - 1; ++i) nrow *= array->shape[i];
    BF_ASSERT(array->ndim == 2, BF_STATUS_UNSUPPORTED_STRIDE);
    return bfMemset2D(array->data, array->strides[0], array->space, value,
                      (BFsize)row_bytes, (BFsize)nrow);
}

}  // extern "C"
