// bifrost_amd: per-thread HIP stream + device context.
// Model follows the reference contract (src/cuda.cpp:34-51): each host
// thread has a thread_local stream (default stream 0); every op enqueues
// async on the calling thread's stream; sync is explicit.
#pragma once

#include <hip/hip_runtime.h>

namespace bfamd {

hipStream_t& thread_stream();

}  // namespace bfamd
