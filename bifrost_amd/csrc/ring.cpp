// bifrost_amd: block-to-block ring buffer.
// ABI: reference src/bifrost/ring.h:74-227.  Re-designed implementation
// (not a translation): one mutex + condvar per ring; absolute monotonic
// byte offsets; a ghost region of `contiguous_bytes` duplicated past the
// end of the buffer so spans that cross the wrap stay contiguous (the
// reference's ghost-region concept, ring_impl.cpp:120-193, re-derived).
// Space-aware: the buffer may live in HIP device memory; ghost fix-up
// copies go through bfMemcpy on the calling thread's stream.
//
// Multi-ringlet (round 2): a ring with nringlet N holds N parallel
// lanes; physical layout is N slabs of (capacity + ghost) bytes, so a
// span's data pointer is ringlet 0's window and the ringlet stride is
// capacity + ghost (what ring.py/ring2.py build their views from,
// reference src/bifrost/ring.h:114 semantics).  Changing nringlet is
// only allowed while the ring is empty (no data, no sequences).

#include <bifrost/memory.h>
#include <bifrost/ring.h>

#include <hip/hip_runtime.h>

#include <algorithm>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <vector>

#include "hipctx.hpp"
#include "status.hpp"

namespace {

struct Sequence {
    std::string name;
    BFoffset time_tag = 0;
    std::vector<char> header;
    BFsize nringlet = 1;
    BFoffset begin = 0;
    BFoffset end = (BFoffset)-1;
    bool ended = false;
    uint64_t id = 0;
};

}  // namespace

struct BFrsequence_impl;

struct BFring_impl {
    std::string name;
    BFspace space;
    int core = -1;

    mutable std::mutex mutex;
    std::condition_variable cv;

    char* buf = nullptr;
    BFsize capacity = 0;   // bytes per ringlet (logical window)
    BFsize ghost = 0;      // contiguous-span bytes duplicated past the end
    BFsize nringlet = 1;   // parallel lanes; slab stride = capacity + ghost

    bool writing_begun = false;
    bool writing_ended = false;

    BFoffset reserve_head = 0;  // frontier of reserved (uncommitted) bytes
    BFoffset head = 0;          // frontier of committed bytes

    // Open-span accounting so resize can wait for quiescence instead of
    // reallocating under a live span pointer (reference
    // ring_impl.cpp:73-80,515-524 RingReallocLock semantics).
    BFsize nwrite_open = 0;
    BFsize nread_open = 0;
    BFsize nrealloc_pending = 0;

    std::deque<std::shared_ptr<Sequence>> sequences;
    uint64_t next_seq_id = 0;

    std::set<BFrsequence_impl*> guaranteed_readers;

    ~BFring_impl() {
        if (buf) bfFree(buf, space);
    }

    BFoffset tail() const {
        return head > capacity ? head - capacity : 0;
    }
    BFoffset guarded_tail() const;  // min guard over guaranteed readers

    char* ptr_at(BFoffset abs) const {
        return buf + (capacity ? (abs % capacity) : 0);
    }

    // Duplicate buffer-start bytes into the ghost area so that a span
    // covering [abs, abs+size) with (abs % capacity) + size > capacity
    // reads valid data.  Caller holds the mutex.
    BFstatus refresh_ghost(BFoffset abs, BFsize size) {
        BFsize pos = abs % capacity;
        if (pos + size <= capacity) return BF_STATUS_SUCCESS;
        BFsize overhang = pos + size - capacity;
        BFsize slab = capacity + ghost;
        for (BFsize r = 0; r < nringlet; ++r) {
            char* base = buf + r * slab;
            BFstatus st = bfMemcpy(base + capacity, space, base, space,
                                   overhang);
            if (st != BF_STATUS_SUCCESS) return st;
        }
        return ghost_fence();
    }
    // Propagate bytes written into the ghost area back to the buffer start
    // after a wrapped write span commits.  Caller holds the mutex.
    BFstatus flush_ghost(BFoffset abs, BFsize size) {
        BFsize pos = abs % capacity;
        if (pos + size <= capacity) return BF_STATUS_SUCCESS;
        BFsize overhang = pos + size - capacity;
        BFsize slab = capacity + ghost;
        for (BFsize r = 0; r < nringlet; ++r) {
            char* base = buf + r * slab;
            BFstatus st = bfMemcpy(base, space, base + capacity, space,
                                   overhang);
            if (st != BF_STATUS_SUCCESS) return st;
        }
        return ghost_fence();
    }
    // Ghost fix-up copies on device-space rings are enqueued async on the
    // CALLING thread's HIP stream; a peer thread with a different stream
    // (bfStreamSet / torch-stream interop) could otherwise observe the span
    // before the copy lands.  Synchronize before publishing.
    BFstatus ghost_fence() const {
        if (space == BF_SPACE_CUDA || space == BF_SPACE_CUDA_MANAGED ||
            space == BF_SPACE_CUDA_HOST) {
            if (hipStreamSynchronize(bfamd::thread_stream()) != hipSuccess)
                return BF_STATUS_DEVICE_ERROR;
        }
        return BF_STATUS_SUCCESS;
    }
};

struct BFwsequence_impl {
    BFring_impl* ring;
    std::shared_ptr<Sequence> seq;
};

struct BFrsequence_impl {
    BFring_impl* ring;
    std::shared_ptr<Sequence> seq;
    bool guarantee = false;
    BFoffset guard = 0;  // absolute offset this reader still needs
    // Spans keep their reader alive across an early Close (round-2 fix:
    // an abandoned read generator finalizing after its sequence closed
    // released through a freed rseq — use-after-free).
    int nspan_open = 0;
    bool closed = false;
};

// Common span header: BFspan is either a write or a read span; the shared
// prefix lets the Get* helpers serve both (discriminated by is_write).
struct BFspan_impl {
    bool is_write;
    BFring_impl* ring;
    BFoffset begin;  // absolute ring offset
    BFsize size;
};

struct BFwspan_impl : BFspan_impl {};

struct BFrspan_impl : BFspan_impl {
    BFrsequence_impl* rseq;
    BFsize size_overwritten;
};

BFoffset BFring_impl::guarded_tail() const {
    BFoffset t = (BFoffset)-1;
    for (auto* r : guaranteed_readers) t = std::min(t, r->guard);
    if (t == (BFoffset)-1) return head;  // no guards: nothing to protect
    return t;
}

extern "C" {

BFstatus bfRingCreate(BFring* ring, const char* name, BFspace space) {
    BF_ASSERT(ring && name, BF_STATUS_INVALID_POINTER);
    BF_TRY_RETURN({
        auto* r = new BFring_impl();
        r->name = name;
        r->space = space;
        *ring = r;
    });
}

BFstatus bfRingDestroy(BFring ring) {
    BF_ASSERT(ring, BF_STATUS_INVALID_HANDLE);
    delete ring;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingResize(BFring ring, BFsize contiguous_bytes,
                      BFsize capacity_bytes, BFsize nringlet) {
    BF_ASSERT(ring, BF_STATUS_INVALID_HANDLE);
    BF_ASSERT(nringlet >= 1, BF_STATUS_INVALID_ARGUMENT);
    BF_TRY_RETURN({
        std::unique_lock<std::mutex> lk(ring->mutex);
        // Growing/shrinking the ringlet count is only defined while the
        // ring holds no data (re-laning live bytes has no meaning).
        BF_THROW_IF(nringlet != ring->nringlet &&
                        (ring->head > 0 || !ring->sequences.empty()),
                    BF_STATUS_UNSUPPORTED_SHAPE);
        BFsize new_nringlet = nringlet;
        BFsize new_ghost = std::max(ring->ghost, contiguous_bytes);
        BFsize new_cap = std::max(ring->capacity,
                                  std::max(capacity_bytes, new_ghost));
        if (new_ghost == ring->ghost && new_cap == ring->capacity &&
            new_nringlet == ring->nringlet)
            return BF_STATUS_SUCCESS;
        // Wait until no span is open: live spans hold raw pointers into the
        // old buffer, so reallocating under them is a use-after-free.  New
        // reserve/acquire calls block while nrealloc_pending > 0.
        ++ring->nrealloc_pending;
        ring->cv.wait(lk, [&] {
            return ring->nwrite_open == 0 && ring->nread_open == 0;
        });
        --ring->nrealloc_pending;
        // Re-check: another resize may have satisfied the request meanwhile.
        new_ghost = std::max(ring->ghost, contiguous_bytes);
        new_cap = std::max(ring->capacity,
                           std::max(capacity_bytes, new_ghost));
        if (new_ghost == ring->ghost && new_cap == ring->capacity &&
            new_nringlet == ring->nringlet) {
            ring->cv.notify_all();
            return BF_STATUS_SUCCESS;
        }
        char* new_buf = nullptr;
        BF_THROW_IF(bfMalloc((void**)&new_buf,
                             new_nringlet * (new_cap + new_ghost),
                             ring->space) != BF_STATUS_SUCCESS,
                    BF_STATUS_MEM_ALLOC_FAILED);
        if (ring->buf && ring->head > 0) {
            // Re-place live bytes [tail, head) at their new positions,
            // per ringlet (nringlet unchanged here by the gate above).
            BFsize old_slab = ring->capacity + ring->ghost;
            BFsize new_slab = new_cap + new_ghost;
            for (BFsize rl = 0; rl < ring->nringlet; ++rl) {
                BFoffset t = ring->tail();
                BFoffset h = ring->head;
                // Chunks contiguous in BOTH old and new layout.
                BFoffset o = t;
                while (o < h) {
                    BFsize old_pos = o % ring->capacity;
                    BFsize new_pos = o % new_cap;
                    BFsize n = std::min((BFsize)(h - o),
                                        std::min(ring->capacity - old_pos,
                                                 new_cap - new_pos));
                    bfMemcpy(new_buf + rl * new_slab + new_pos, ring->space,
                             ring->buf + rl * old_slab + old_pos,
                             ring->space, n);
                    o += n;
                }
            }
        }
        if (ring->buf) bfFree(ring->buf, ring->space);
        ring->buf = new_buf;
        ring->capacity = new_cap;
        ring->ghost = new_ghost;
        ring->nringlet = new_nringlet;
        ring->cv.notify_all();
    });
}

BFstatus bfRingGetName(BFring ring, const char** name) {
    BF_ASSERT(ring && name, BF_STATUS_INVALID_POINTER);
    *name = ring->name.c_str();
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingGetSpace(BFring ring, BFspace* space) {
    BF_ASSERT(ring && space, BF_STATUS_INVALID_POINTER);
    *space = ring->space;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingSetAffinity(BFring ring, int core) {
    BF_ASSERT(ring, BF_STATUS_INVALID_HANDLE);
    ring->core = core;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingGetAffinity(BFring ring, int* core) {
    BF_ASSERT(ring && core, BF_STATUS_INVALID_POINTER);
    *core = ring->core;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingLock(BFring ring) {
    BF_ASSERT(ring, BF_STATUS_INVALID_HANDLE);
    ring->mutex.lock();
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingUnlock(BFring ring) {
    BF_ASSERT(ring, BF_STATUS_INVALID_HANDLE);
    ring->mutex.unlock();
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingLockedGetData(BFring ring, void** data) {
    BF_ASSERT(ring && data, BF_STATUS_INVALID_POINTER);
    *data = ring->buf;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingLockedGetContiguousSpan(BFring ring, BFsize* val) {
    BF_ASSERT(ring && val, BF_STATUS_INVALID_POINTER);
    *val = ring->ghost;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingLockedGetTotalSpan(BFring ring, BFsize* val) {
    BF_ASSERT(ring && val, BF_STATUS_INVALID_POINTER);
    *val = ring->capacity;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingLockedGetNRinglet(BFring ring, BFsize* val) {
    BF_ASSERT(ring && val, BF_STATUS_INVALID_POINTER);
    *val = 1;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingLockedGetStride(BFring ring, BFsize* val) {
    BF_ASSERT(ring && val, BF_STATUS_INVALID_POINTER);
    *val = ring->capacity;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingBeginWriting(BFring ring) {
    BF_ASSERT(ring, BF_STATUS_INVALID_HANDLE);
    std::lock_guard<std::mutex> lk(ring->mutex);
    BF_ASSERT(!ring->writing_ended, BF_STATUS_INVALID_STATE);
    ring->writing_begun = true;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingEndWriting(BFring ring) {
    BF_ASSERT(ring, BF_STATUS_INVALID_HANDLE);
    std::lock_guard<std::mutex> lk(ring->mutex);
    ring->writing_ended = true;
    ring->cv.notify_all();
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingWritingEnded(BFring ring, BFbool* writing_ended) {
    BF_ASSERT(ring && writing_ended, BF_STATUS_INVALID_POINTER);
    std::lock_guard<std::mutex> lk(ring->mutex);
    *writing_ended = ring->writing_ended;
    return BF_STATUS_SUCCESS;
}

/* ------------------------------ sequences ----------------------------- */

BFstatus bfRingSequenceBegin(BFwsequence* sequence, BFring ring,
                             const char* name, BFoffset time_tag,
                             BFsize header_size, const void* header,
                             BFsize nringlet, BFoffset offset_from_head) {
    BF_ASSERT(sequence && ring, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(nringlet >= 1, BF_STATUS_INVALID_ARGUMENT);
    // The ring's physical lanes must cover the sequence's (resize first).
    BF_ASSERT(nringlet <= ring->nringlet, BF_STATUS_UNSUPPORTED_SHAPE);
    BF_TRY_RETURN({
        std::lock_guard<std::mutex> lk(ring->mutex);
        auto seq = std::make_shared<Sequence>();
        seq->name = name ? name : "";
        seq->time_tag = time_tag;
        if (header_size) {
            seq->header.assign((const char*)header,
                               (const char*)header + header_size);
        }
        seq->nringlet = nringlet;
        seq->begin = ring->head + offset_from_head;
        seq->id = ring->next_seq_id++;
        ring->sequences.push_back(seq);
        auto* ws = new BFwsequence_impl{ring, seq};
        *sequence = ws;
        ring->cv.notify_all();
    });
}

BFstatus bfRingSequenceEnd(BFwsequence sequence, BFoffset offset_from_head) {
    BF_ASSERT(sequence, BF_STATUS_INVALID_HANDLE);
    BFring_impl* ring = sequence->ring;
    {
        std::lock_guard<std::mutex> lk(ring->mutex);
        sequence->seq->end = ring->head + offset_from_head;
        sequence->seq->ended = true;
        ring->cv.notify_all();
    }
    delete sequence;
    return BF_STATUS_SUCCESS;
}

namespace {

// Open helper.  which: 0=by-name, 1=at-time, 2=latest, 3=earliest.
BFstatus open_sequence(BFrsequence* sequence, BFring ring, int which,
                       const char* name, BFoffset time_tag, BFbool guarantee) {
    BF_ASSERT(sequence && ring, BF_STATUS_INVALID_POINTER);
    std::unique_lock<std::mutex> lk(ring->mutex);
    std::shared_ptr<Sequence> found;
    for (;;) {
        if (which == 0) {
            for (auto& s : ring->sequences)
                if (s->name == name) { found = s; break; }
        } else if (which == 1) {
            // last sequence whose time_tag <= requested
            for (auto& s : ring->sequences) {
                if (s->time_tag <= time_tag) found = s;
            }
        } else if (which == 2) {
            if (!ring->sequences.empty()) found = ring->sequences.back();
        } else {
            // earliest not yet fully overwritten
            for (auto& s : ring->sequences) {
                bool gone = s->ended && s->end <= ring->tail();
                if (!gone) { found = s; break; }
            }
        }
        if (found) break;
        if (ring->writing_ended) return BF_STATUS_END_OF_DATA;
        ring->cv.wait(lk);
    }
    auto* rs = new BFrsequence_impl();
    rs->ring = ring;
    rs->seq = found;
    rs->guarantee = guarantee;
    rs->guard = std::max(found->begin, ring->tail());
    if (guarantee) {
        ring->guaranteed_readers.insert(rs);
    }
    *sequence = rs;
    return BF_STATUS_SUCCESS;
}

}  // namespace

BFstatus bfRingSequenceOpen(BFrsequence* sequence, BFring ring,
                            const char* name, BFbool guarantee) {
    BF_ASSERT(name, BF_STATUS_INVALID_POINTER);
    return open_sequence(sequence, ring, 0, name, 0, guarantee);
}
BFstatus bfRingSequenceOpenAt(BFrsequence* sequence, BFring ring,
                              BFoffset time_tag, BFbool guarantee) {
    return open_sequence(sequence, ring, 1, nullptr, time_tag, guarantee);
}
BFstatus bfRingSequenceOpenLatest(BFrsequence* sequence, BFring ring,
                                  BFbool guarantee) {
    return open_sequence(sequence, ring, 2, nullptr, 0, guarantee);
}
BFstatus bfRingSequenceOpenEarliest(BFrsequence* sequence, BFring ring,
                                    BFbool guarantee) {
    return open_sequence(sequence, ring, 3, nullptr, 0, guarantee);
}

BFstatus bfRingSequenceNext(BFrsequence sequence) {
    BF_ASSERT(sequence, BF_STATUS_INVALID_HANDLE);
    BFring_impl* ring = sequence->ring;
    std::unique_lock<std::mutex> lk(ring->mutex);
    uint64_t want_id = sequence->seq->id + 1;
    for (;;) {
        std::shared_ptr<Sequence> next;
        for (auto& s : ring->sequences)
            if (s->id == want_id) { next = s; break; }
        if (next) {
            sequence->seq = next;
            sequence->guard = std::max(next->begin, ring->tail());
            ring->cv.notify_all();
            return BF_STATUS_SUCCESS;
        }
        if (ring->writing_ended) return BF_STATUS_END_OF_DATA;
        ring->cv.wait(lk);
    }
}

BFstatus bfRingSequenceClose(BFrsequence sequence) {
    BF_ASSERT(sequence, BF_STATUS_INVALID_HANDLE);
    BFring_impl* ring = sequence->ring;
    bool defer;
    {
        std::lock_guard<std::mutex> lk(ring->mutex);
        ring->guaranteed_readers.erase(sequence);
        ring->cv.notify_all();
        defer = sequence->nspan_open > 0;
        if (defer) sequence->closed = true;  // last span release frees
    }
    if (!defer) delete sequence;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingSequenceGetRing(BFsequence sequence, BFring* ring) {
    BF_ASSERT(sequence && ring, BF_STATUS_INVALID_POINTER);
    *ring = ((BFrsequence_impl*)sequence)->ring;
    return BF_STATUS_SUCCESS;
}
// Note: BFsequence handles are BFrsequence_impl*/BFwsequence_impl* cast to a
// common view; both structs begin with {ring, seq} so this is well-defined.
static Sequence* seq_of(BFsequence s) {
    return ((BFrsequence_impl*)s)->seq.get();
}
BFstatus bfRingSequenceGetName(BFsequence sequence, const char** name) {
    BF_ASSERT(sequence && name, BF_STATUS_INVALID_POINTER);
    *name = seq_of(sequence)->name.c_str();
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSequenceGetTimeTag(BFsequence sequence, BFoffset* time_tag) {
    BF_ASSERT(sequence && time_tag, BF_STATUS_INVALID_POINTER);
    *time_tag = seq_of(sequence)->time_tag;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSequenceGetHeader(BFsequence sequence, const void** hdr) {
    BF_ASSERT(sequence && hdr, BF_STATUS_INVALID_POINTER);
    *hdr = seq_of(sequence)->header.data();
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSequenceGetHeaderSize(BFsequence sequence, BFsize* size) {
    BF_ASSERT(sequence && size, BF_STATUS_INVALID_POINTER);
    *size = seq_of(sequence)->header.size();
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSequenceGetNRinglet(BFsequence sequence, BFsize* nringlet) {
    BF_ASSERT(sequence && nringlet, BF_STATUS_INVALID_POINTER);
    *nringlet = seq_of(sequence)->nringlet;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSequenceGetInfo(BFsequence sequence, BFsequence_info* info) {
    BF_ASSERT(sequence && info, BF_STATUS_INVALID_POINTER);
    Sequence* s = seq_of(sequence);
    info->ring = ((BFrsequence_impl*)sequence)->ring;
    info->name = s->name.c_str();
    info->time_tag = s->time_tag;
    info->header = s->header.data();
    info->header_size = s->header.size();
    info->nringlet = s->nringlet;
    return BF_STATUS_SUCCESS;
}

/* -------------------------------- spans -------------------------------- */

BFstatus bfRingSpanReserve(BFwspan* span, BFring ring,
                           BFsize size, BFbool nonblocking) {
    BF_ASSERT(span && ring, BF_STATUS_INVALID_POINTER);
    std::unique_lock<std::mutex> lk(ring->mutex);
    BF_ASSERT(ring->buf, BF_STATUS_INVALID_STATE);
    BF_ASSERT(size <= ring->ghost, BF_STATUS_INVALID_ARGUMENT);
    // Block while committing would overwrite data a guaranteed reader still
    // needs.  Non-guaranteed data is overwritten freely (readers detect it).
    for (;;) {
        if (ring->nrealloc_pending) {
            // A resize is waiting for quiescence; don't open new spans.
            if (nonblocking) return BF_STATUS_WOULD_BLOCK;
            ring->cv.wait(lk);
            continue;
        }
        BFoffset gt = ring->guarded_tail();
        if (ring->reserve_head + size <= gt + ring->capacity) break;
        if (ring->writing_ended) return BF_STATUS_INVALID_STATE;
        if (nonblocking) return BF_STATUS_WOULD_BLOCK;
        ring->cv.wait(lk);
    }
    auto* ws = new BFwspan_impl();
    ws->is_write = true;
    ws->ring = ring;
    ws->begin = ring->reserve_head;
    ws->size = size;
    ring->reserve_head += size;
    // Make sure a wrapped write span reflects current buffer-start contents
    // (partial commits elsewhere could otherwise be clobbered by flush).
    BFstatus st = ring->refresh_ghost(ws->begin, size);
    if (st != BF_STATUS_SUCCESS) {
        // Roll back the reservation so the window isn't leaked until an
        // unrelated commit resets reserve_head.
        ring->reserve_head -= size;
        delete ws;
        ring->cv.notify_all();
        return st;
    }
    ++ring->nwrite_open;
    *span = ws;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingSpanCommit(BFwspan span, BFsize size) {
    BF_ASSERT(span, BF_STATUS_INVALID_HANDLE);
    BF_ASSERT(size <= span->size, BF_STATUS_INVALID_ARGUMENT);
    BFring_impl* ring = span->ring;
    {
        std::lock_guard<std::mutex> lk(ring->mutex);
        BFstatus st = ring->flush_ghost(span->begin, size);
        if (st != BF_STATUS_SUCCESS) {
            --ring->nwrite_open;
            ring->cv.notify_all();
            delete span;
            return st;
        }
        ring->head = span->begin + size;
        ring->reserve_head = ring->head;
        --ring->nwrite_open;
        ring->cv.notify_all();
    }
    delete span;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingSpanAcquire(BFrspan* span, BFrsequence sequence,
                           BFoffset offset, BFsize size) {
    BF_ASSERT(span && sequence, BF_STATUS_INVALID_POINTER);
    BFring_impl* ring = sequence->ring;
    std::unique_lock<std::mutex> lk(ring->mutex);
    BF_ASSERT(ring->buf, BF_STATUS_INVALID_STATE);
    BF_ASSERT(size <= ring->ghost, BF_STATUS_INVALID_ARGUMENT);
    Sequence* seq = sequence->seq.get();
    // Reference semantics (src/ring_impl.cpp:633-701): return whatever
    // part of the REQUESTED window [req_begin, req_begin+size) survives —
    // begin = max(req_begin, tail), possibly a zero-length span when the
    // window was fully overwritten.  NEVER jump beyond the request: the
    // caller's frame-offset bookkeeping (ring2.read 'offset += stride')
    // depends on each acquire covering exactly its window, and the
    // pipeline's _on_skip/zero-span paths handle the rest.  (The earlier
    // fast-forward-to-tail behaviour re-delivered fresh data for stale
    // offsets: an unguaranteed reader chain then REPLAYS the stream ~2x
    // per hop — exponential in pipeline depth.)
    BFoffset req_begin = seq->begin + offset;
    BFoffset req_end = req_begin + size;
    if (sequence->guarantee && req_begin > sequence->guard) {
        // move the guarantee forward so writers can progress
        sequence->guard = req_begin;
        ring->cv.notify_all();
    }
    for (;;) {
        if (ring->nrealloc_pending) {
            // A resize is waiting for quiescence; don't open new spans.
            ring->cv.wait(lk);
            continue;
        }
        if (seq->ended && req_begin >= seq->end) return BF_STATUS_END_OF_DATA;
        if (ring->writing_ended && !seq->ended && req_begin >= ring->head)
            return BF_STATUS_END_OF_DATA;
        BFoffset avail_begin = std::max<BFoffset>(req_begin, ring->tail());
        bool finished = seq->ended ||
                        (ring->writing_ended && !seq->ended);
        if ((BFdelta)(ring->head - avail_begin) >=
                (BFdelta)(req_end - avail_begin) ||
            finished) {
            BFoffset begin = std::max<BFoffset>(req_begin, ring->tail());
            BFdelta ssize = (BFdelta)(req_end - begin);
            if (ssize < 0) ssize = 0;
            if (seq->ended) {
                if (begin >= seq->end) return BF_STATUS_END_OF_DATA;
                ssize = std::min<BFdelta>(ssize, (BFdelta)(seq->end - begin));
            } else if (ring->writing_ended) {
                if (begin >= ring->head) return BF_STATUS_END_OF_DATA;
                ssize = std::min<BFdelta>(ssize,
                                          (BFdelta)(ring->head - begin));
            } else if ((BFdelta)(ring->head - begin) < ssize) {
                // window partially overwritten but head not there yet
                ring->cv.wait(lk);
                continue;
            }
            if (ssize > 0) {
                BFstatus st = ring->refresh_ghost(begin, (BFsize)ssize);
                if (st != BF_STATUS_SUCCESS) return st;
            }
            sequence->guard = begin;
            ++ring->nread_open;
            ++sequence->nspan_open;
            auto* rs = new BFrspan_impl();
            rs->is_write = false;
            rs->ring = ring;
            rs->rseq = sequence;
            rs->begin = begin;
            rs->size = (BFsize)ssize;
            rs->size_overwritten = 0;
            *span = rs;
            return BF_STATUS_SUCCESS;
        }
        ring->cv.wait(lk);
    }
}

BFstatus bfRingSpanRelease(BFrspan span) {
    BF_ASSERT(span, BF_STATUS_INVALID_HANDLE);
    BFrsequence_impl* rseq = span->rseq;
    BFring_impl* ring = rseq->ring;
    bool free_rseq;
    {
        std::lock_guard<std::mutex> lk(ring->mutex);
        // Advance the reader's guard past this span (sequential-gulp model).
        rseq->guard = std::max(rseq->guard, span->begin + span->size);
        --ring->nread_open;
        free_rseq = (--rseq->nspan_open == 0) && rseq->closed;
        ring->cv.notify_all();
    }
    if (free_rseq) delete rseq;
    delete span;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingSpanGetSizeOverwritten(BFrspan span, BFsize* val) {
    BF_ASSERT(span && val, BF_STATUS_INVALID_POINTER);
    BFring_impl* ring = span->rseq->ring;
    std::lock_guard<std::mutex> lk(ring->mutex);
    if (span->rseq->guarantee) {
        *val = 0;
    } else {
        BFoffset t = ring->tail();
        *val = t > span->begin
                   ? (BFsize)std::min<BFoffset>(t - span->begin, span->size)
                   : 0;
    }
    return BF_STATUS_SUCCESS;
}

/* Any-span getters (span handles carry a tagged common header). */

BFstatus bfRingSpanGetInfo(BFspan span, BFspan_info* info) {
    BF_ASSERT(span && info, BF_STATUS_INVALID_POINTER);
    auto* sp = (BFspan_impl*)span;
    BFring_impl* ring = sp->ring;
    std::lock_guard<std::mutex> lk(ring->mutex);
    info->ring = ring;
    info->data = ring->ptr_at(sp->begin);
    info->size = sp->size;
    // ringlet stride = the per-lane slab (capacity + ghost); read spans
    // report their sequence's lane count, write spans the ring's.
    info->stride = ring->capacity + ring->ghost;
    info->nringlet = sp->is_write ? ring->nringlet
                                  : ((BFrspan_impl*)sp)->rseq->seq->nringlet;
    if (sp->is_write) {
        info->offset = (BFsize)sp->begin;
    } else {
        auto* rs = (BFrspan_impl*)sp;
        info->offset = (BFsize)(sp->begin - rs->rseq->seq->begin);
    }
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingSpanGetRing(BFspan span, BFring* out) {
    BF_ASSERT(span && out, BF_STATUS_INVALID_POINTER);
    *out = ((BFspan_impl*)span)->ring;
    return BF_STATUS_SUCCESS;
}

BFstatus bfRingSpanGetData(BFspan span, void** data) {
    BFspan_info info;
    BF_CHECK(bfRingSpanGetInfo(span, &info));
    *data = info.data;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSpanGetSize(BFspan span, BFsize* val) {
    BFspan_info info;
    BF_CHECK(bfRingSpanGetInfo(span, &info));
    *val = info.size;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSpanGetStride(BFspan span, BFsize* val) {
    BFspan_info info;
    BF_CHECK(bfRingSpanGetInfo(span, &info));
    *val = info.stride;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSpanGetOffset(BFspan span, BFsize* val) {
    BFspan_info info;
    BF_CHECK(bfRingSpanGetInfo(span, &info));
    *val = info.offset;
    return BF_STATUS_SUCCESS;
}
BFstatus bfRingSpanGetNRinglet(BFspan span, BFsize* val) {
    BFspan_info info;
    BF_CHECK(bfRingSpanGetInfo(span, &info));
    *val = info.nringlet;
    return BF_STATUS_SUCCESS;
}

}  // extern "C"
