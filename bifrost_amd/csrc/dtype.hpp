// bifrost_amd: BFdtype/BFarray helpers.  Size semantics follow the
// reference contract (src/utils.hpp:44-77,255-277): nbit counts BOTH
// components of a complex type; sub-byte arrays carry their last shape dim
// in elements while strides describe the byte-packed layout.
#pragma once

#include <bifrost/array.h>

#include "status.hpp"

namespace bfamd {

inline bool dtype_is_complex(BFdtype dt) { return dt & BF_DTYPE_COMPLEX_BIT; }
inline bool dtype_is_float(BFdtype dt) {
    return (dt & BF_DTYPE_TYPE_BITS) == BF_DTYPE_FLOAT_TYPE;
}
inline bool dtype_is_signed_int(BFdtype dt) {
    return (dt & BF_DTYPE_TYPE_BITS) == BF_DTYPE_INT_TYPE;
}
inline int dtype_nbit(BFdtype dt) {
    int veclen = ((dt & BF_DTYPE_VECTOR_BITS) >> BF_DTYPE_VECTOR_BIT0) + 1;
    return (dt & BF_DTYPE_NBIT_BITS) * (dtype_is_complex(dt) ? 2 : 1) * veclen;
}
inline int dtype_nbyte(BFdtype dt) { return dtype_nbit(dt) / 8; }

inline BFsize capacity_bytes(const BFarray* a) {
    // Outermost extent; assumes strides[0] is the largest (row-major-ish).
    return (BFsize)a->strides[0] * (BFsize)a->shape[0];
}

inline bool is_contiguous(const BFarray* a) {
    BFsize logical = dtype_nbyte(a->dtype);
    for (int d = 0; d < a->ndim; ++d) logical *= a->shape[d];
    return a->ndim > 0 && logical == capacity_bytes(a);
}

inline BFsize num_contiguous_elements(const BFarray* a) {
    return capacity_bytes(a) / dtype_nbyte(a->dtype);
}

inline bool shapes_equal(const BFarray* a, const BFarray* b) {
    if (a->ndim != b->ndim) return false;
    for (int d = 0; d < a->ndim; ++d)
        if (a->shape[d] != b->shape[d]) return false;
    return true;
}

inline long num_elements(const BFarray* a) {
    long n = 1;
    for (int d = 0; d < a->ndim; ++d) n *= a->shape[d];
    return n;
}

inline bool space_on_device(BFspace s) {
    return s == BF_SPACE_CUDA || s == BF_SPACE_CUDA_MANAGED;
}
inline bool space_host_accessible(BFspace s) {
    return s == BF_SPACE_SYSTEM || s == BF_SPACE_CUDA_HOST ||
           s == BF_SPACE_CUDA_MANAGED;
}
inline bool space_device_accessible(BFspace s) {
    return s == BF_SPACE_CUDA || s == BF_SPACE_CUDA_HOST ||
           s == BF_SPACE_CUDA_MANAGED;
}

// Dim-merging semantics of the reference dispatch (utils.hpp:348-366,
// 397-408): a set bit at dim d in keep_mask CLOSES the merge group at d;
// padded_dims_mask marks boundaries that must not merge (bit d-1 set when
// strides[d]*shape[d] != strides[d-1]).  All arrays of one matmul call are
// flattened with the SAME mask so their flattened ndims agree.
inline unsigned long padded_dims_mask(const BFarray* a) {
    unsigned long mask = 0;
    for (int d = 1; d < a->ndim; ++d) {
        bool padded = a->strides[d] * a->shape[d] != a->strides[d - 1];
        mask |= ((unsigned long)padded) << (d - 1);
    }
    return mask;
}

inline void flatten_dims(const BFarray* in, BFarray* out,
                         unsigned long keep_mask) {
    *out = *in;
    int od = 0;
    long osize = 1;
    for (int d = 0; d < in->ndim; ++d) {
        osize *= in->shape[d];
        bool last = d == in->ndim - 1;
        bool keep = (keep_mask >> d) & 1;
        if (last || keep) {
            out->shape[od] = osize;
            out->strides[od] = in->strides[d];
            osize = 1;
            ++od;
        }
    }
    out->ndim = od;
}

}  // namespace bfamd
