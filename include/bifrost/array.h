/* bifrost_amd: the BFarray POD and dtype encoding.
 *
 * The BFdtype bitfield values and BFarray struct layout reproduce
 * reference src/bifrost/array.h:41-132 exactly (ABI contract: 8-dim
 * max, element shapes, BYTE strides, conjugated flag, caller owns data).
 */
#ifndef BFAMD_ARRAY_H_
#define BFAMD_ARRAY_H_

#include <bifrost/common.h>
#include <bifrost/memory.h>

#ifdef __cplusplus
extern "C" {
#endif

enum { BF_MAX_DIMS = 8 };

typedef enum BFdtype_ {
    BF_DTYPE_NBIT_BITS    = 0x0000FF,
    BF_DTYPE_TYPE_BITS    = 0x000F00,
    BF_DTYPE_VECTOR_BITS  = 0x0FF000,
    BF_DTYPE_VECTOR_BIT0  = 12,
    BF_DTYPE_COMPLEX_BIT  = 0x100000,

    BF_DTYPE_INT_TYPE     = 0x0000,
    BF_DTYPE_UINT_TYPE    = 0x0100,
    BF_DTYPE_FLOAT_TYPE   = 0x0200,
    BF_DTYPE_STRING_TYPE  = 0x0300,
    BF_DTYPE_STORAGE_TYPE = 0x0400,

    BF_DTYPE_I1  =  1 | BF_DTYPE_INT_TYPE,
    BF_DTYPE_I2  =  2 | BF_DTYPE_INT_TYPE,
    BF_DTYPE_I4  =  4 | BF_DTYPE_INT_TYPE,
    BF_DTYPE_I8  =  8 | BF_DTYPE_INT_TYPE,
    BF_DTYPE_I16 = 16 | BF_DTYPE_INT_TYPE,
    BF_DTYPE_I32 = 32 | BF_DTYPE_INT_TYPE,
    BF_DTYPE_I64 = 64 | BF_DTYPE_INT_TYPE,

    BF_DTYPE_U1  =  1 | BF_DTYPE_UINT_TYPE,
    BF_DTYPE_U2  =  2 | BF_DTYPE_UINT_TYPE,
    BF_DTYPE_U4  =  4 | BF_DTYPE_UINT_TYPE,
    BF_DTYPE_U8  =  8 | BF_DTYPE_UINT_TYPE,
    BF_DTYPE_U16 = 16 | BF_DTYPE_UINT_TYPE,
    BF_DTYPE_U32 = 32 | BF_DTYPE_UINT_TYPE,
    BF_DTYPE_U64 = 64 | BF_DTYPE_UINT_TYPE,

    BF_DTYPE_F16 = 16 | BF_DTYPE_FLOAT_TYPE,
    BF_DTYPE_F32 = 32 | BF_DTYPE_FLOAT_TYPE,
    BF_DTYPE_F64 = 64 | BF_DTYPE_FLOAT_TYPE,

    BF_DTYPE_CI1  =  1 | BF_DTYPE_INT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CI2  =  2 | BF_DTYPE_INT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CI4  =  4 | BF_DTYPE_INT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CI8  =  8 | BF_DTYPE_INT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CI16 = 16 | BF_DTYPE_INT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CI32 = 32 | BF_DTYPE_INT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CI64 = 64 | BF_DTYPE_INT_TYPE | BF_DTYPE_COMPLEX_BIT,

    BF_DTYPE_CF16 = 16 | BF_DTYPE_FLOAT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CF32 = 32 | BF_DTYPE_FLOAT_TYPE | BF_DTYPE_COMPLEX_BIT,
    BF_DTYPE_CF64 = 64 | BF_DTYPE_FLOAT_TYPE | BF_DTYPE_COMPLEX_BIT
} BFdtype;

typedef struct BFarray_ {
    void*    data;
    BFspace  space;
    BFdtype  dtype;
    int      ndim;
    long     shape[BF_MAX_DIMS];   /* elements */
    long     strides[BF_MAX_DIMS]; /* bytes    */
    BFbool   immutable;
    BFbool   big_endian;
    BFbool   conjugated;
} BFarray;

BFstatus bfArrayMalloc(BFarray* array);
BFstatus bfArrayFree(const BFarray* array);
BFstatus bfArrayCopy(const BFarray* dst, const BFarray* src);
BFstatus bfArrayMemset(const BFarray* array, int value);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_ARRAY_H_ */
