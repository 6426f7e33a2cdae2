// bifrost_amd: CPU core binding for block threads.
// ABI: reference src/bifrost/affinity.h:40-44.

#include <bifrost/affinity.h>

#include <pthread.h>
#include <sched.h>
#include <unistd.h>

#include "status.hpp"

extern "C" {

BFstatus bfAffinitySetCore(int core) {
    cpu_set_t cs;
    CPU_ZERO(&cs);
    if (core < 0) {
        long n = sysconf(_SC_NPROCESSORS_ONLN);
        for (long c = 0; c < n; ++c) CPU_SET(c, &cs);
    } else {
        CPU_SET(core, &cs);
    }
    int rc = pthread_setaffinity_np(pthread_self(), sizeof(cs), &cs);
    BF_ASSERT(rc == 0, BF_STATUS_INVALID_ARGUMENT);
    return BF_STATUS_SUCCESS;
}

BFstatus bfAffinityGetCore(int* core) {
    BF_ASSERT(core, BF_STATUS_INVALID_POINTER);
    cpu_set_t cs;
    CPU_ZERO(&cs);
    int rc = pthread_getaffinity_np(pthread_self(), sizeof(cs), &cs);
    BF_ASSERT(rc == 0, BF_STATUS_INTERNAL_ERROR);
    int count = CPU_COUNT(&cs);
    long n = sysconf(_SC_NPROCESSORS_ONLN);
    if (count == 1) {
        for (long c = 0; c < n; ++c) {
            if (CPU_ISSET(c, &cs)) { *core = (int)c; return BF_STATUS_SUCCESS; }
        }
    }
    *core = -1;  // unbound or multi-bound
    return BF_STATUS_SUCCESS;
}

BFstatus bfAffinitySetOpenMPCores(BFsize nthread, const int* thread_cores) {
    // OpenMP is not used by this backend's kernels; accept and ignore.
    (void)nthread;
    (void)thread_cores;
    return BF_STATUS_SUCCESS;
}

}  // extern "C"
