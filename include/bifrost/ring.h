/* bifrost_amd: the block-to-block ring buffer (sequences + spans, ghost
 * region for wrap-contiguous gulps, guarantee semantics).
 * ABI identical to reference src/bifrost/ring.h:74-227.
 */
#ifndef BFAMD_RING_H_
#define BFAMD_RING_H_

#include <bifrost/common.h>
#include <bifrost/memory.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct BFring_impl*        BFring;
typedef struct BFsequence_wrapper* BFsequence;
typedef struct BFrsequence_impl*   BFrsequence;
typedef struct BFwsequence_impl*   BFwsequence;
typedef struct BFspan_impl*        BFspan;
typedef struct BFrspan_impl*       BFrspan;
typedef struct BFwspan_impl*       BFwspan;

/* Ring lifecycle */
BFstatus bfRingCreate(BFring* ring, const char* name, BFspace space);
BFstatus bfRingDestroy(BFring ring);
BFstatus bfRingResize(BFring ring,
                      BFsize contiguous_bytes,
                      BFsize capacity_bytes,
                      BFsize nringlet);
BFstatus bfRingGetName(BFring ring, const char** name);
BFstatus bfRingGetSpace(BFring ring, BFspace* space);
BFstatus bfRingSetAffinity(BFring ring, int core);
BFstatus bfRingGetAffinity(BFring ring, int* core);

BFstatus bfRingLock(BFring ring);
BFstatus bfRingUnlock(BFring ring);
BFstatus bfRingLockedGetData(BFring ring, void** data);
BFstatus bfRingLockedGetContiguousSpan(BFring ring, BFsize* val);
BFstatus bfRingLockedGetTotalSpan(BFring ring, BFsize* val);
BFstatus bfRingLockedGetNRinglet(BFring ring, BFsize* val);
BFstatus bfRingLockedGetStride(BFring ring, BFsize* val);

BFstatus bfRingBeginWriting(BFring ring);
BFstatus bfRingEndWriting(BFring ring);
BFstatus bfRingWritingEnded(BFring ring, BFbool* writing_ended);

/* Sequence write */
BFstatus bfRingSequenceBegin(BFwsequence* sequence,
                             BFring       ring,
                             const char*  name,
                             BFoffset     time_tag,
                             BFsize       header_size,
                             const void*  header,
                             BFsize       nringlet,
                             BFoffset     offset_from_head);
BFstatus bfRingSequenceEnd(BFwsequence sequence, BFoffset offset_from_head);

/* Sequence read */
BFstatus bfRingSequenceOpen(BFrsequence* sequence, BFring ring,
                            const char* name, BFbool guarantee);
BFstatus bfRingSequenceOpenAt(BFrsequence* sequence, BFring ring,
                              BFoffset time_tag, BFbool guarantee);
BFstatus bfRingSequenceOpenLatest(BFrsequence* sequence, BFring ring,
                                  BFbool guarantee);
BFstatus bfRingSequenceOpenEarliest(BFrsequence* sequence, BFring ring,
                                    BFbool guarantee);
BFstatus bfRingSequenceNext(BFrsequence sequence);
BFstatus bfRingSequenceClose(BFrsequence sequence);

/* Sequence common */
BFstatus bfRingSequenceGetRing(BFsequence sequence, BFring* ring);
BFstatus bfRingSequenceGetName(BFsequence sequence, const char** name);
BFstatus bfRingSequenceGetTimeTag(BFsequence sequence, BFoffset* time_tag);
BFstatus bfRingSequenceGetHeader(BFsequence sequence, const void** hdr);
BFstatus bfRingSequenceGetHeaderSize(BFsequence sequence, BFsize* size);
BFstatus bfRingSequenceGetNRinglet(BFsequence sequence, BFsize* nringlet);

typedef struct BFsequence_info_ {
    BFring      ring;
    const char* name;
    BFoffset    time_tag;
    const void* header;
    BFsize      header_size;
    BFsize      nringlet;
} BFsequence_info;
BFstatus bfRingSequenceGetInfo(BFsequence sequence, BFsequence_info* sequence_info);

/* Write span */
BFstatus bfRingSpanReserve(BFwspan* span, BFring ring,
                           BFsize size, BFbool nonblocking);
BFstatus bfRingSpanCommit(BFwspan span, BFsize size);

/* Read span */
BFstatus bfRingSpanAcquire(BFrspan* span, BFrsequence sequence,
                           BFoffset offset, BFsize size);
BFstatus bfRingSpanRelease(BFrspan span);
BFstatus bfRingSpanGetSizeOverwritten(BFrspan span, BFsize* val);

/* Any span */
BFstatus bfRingSpanGetRing(BFspan span, BFring* data);
BFstatus bfRingSpanGetData(BFspan span, void** data);
BFstatus bfRingSpanGetSize(BFspan span, BFsize* val);
BFstatus bfRingSpanGetStride(BFspan span, BFsize* val);
BFstatus bfRingSpanGetOffset(BFspan span, BFsize* val);
BFstatus bfRingSpanGetNRinglet(BFspan span, BFsize* val);

typedef struct BFspan_info_ {
    BFring      ring;
    void*       data;
    BFsize      size;
    BFsize      stride;
    BFsize      offset;
    BFsize      nringlet;
} BFspan_info;
BFstatus bfRingSpanGetInfo(BFspan span, BFspan_info* span_info);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_RING_H_ */
