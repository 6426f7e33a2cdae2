// Built-in C-level self-test: `bfTestSuite()` returns the number of
// failed checks (0 = success), callable by any ABI consumer as a
// post-install sanity check.  Mirrors the role of the reference's
// src/testsuite.cpp:189 (bfTestSuite) — the reference checks its
// fileutils; ours exercises the pieces this backend actually ships:
// filesystem helpers, system-space memory, and a full ring
// write/read round trip through the public C API.

#include <bifrost/testsuite.h>
#include <bifrost/common.h>
#include <bifrost/memory.h>
#include <bifrost/ring.h>

#include <sys/stat.h>
#include <unistd.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>

namespace {

#define TS_CHECK(cond)                                                  \
    do {                                                                \
        if (!(cond)) {                                                  \
            std::fprintf(stderr, "testsuite: FAIL: %s @%s:%d\n",        \
                         #cond, __FUNCTION__, __LINE__);                \
            return 1;                                                   \
        }                                                               \
    } while (0)

std::string tmp_root() {
    const char* t = std::getenv("TMPDIR");
    return std::string(t && *t ? t : "/tmp");
}

int test_this_process_exists() {
    // /proc/<pid> must exist for our own pid (the proclog layer keys its
    // directories by pid and reaps dead ones this way).
    std::string p = "/proc/" + std::to_string(getpid());
    struct stat st;
    TS_CHECK(stat(p.c_str(), &st) == 0 && S_ISDIR(st.st_mode));
    return 0;
}

int test_make_then_remove_dir() {
    std::string d = tmp_root() + "/bfamd_ts_dir_" + std::to_string(getpid());
    TS_CHECK(mkdir(d.c_str(), 0777) == 0);
    struct stat st;
    TS_CHECK(stat(d.c_str(), &st) == 0 && S_ISDIR(st.st_mode));
    TS_CHECK(rmdir(d.c_str()) == 0);
    TS_CHECK(stat(d.c_str(), &st) != 0);
    return 0;
}

int test_create_then_remove_file() {
    std::string f = tmp_root() + "/bfamd_ts_file_" + std::to_string(getpid());
    FILE* fp = std::fopen(f.c_str(), "w");
    TS_CHECK(fp != nullptr);
    std::fputs("bifrost_amd", fp);
    std::fclose(fp);
    struct stat st;
    TS_CHECK(stat(f.c_str(), &st) == 0 && st.st_size == 11);
    TS_CHECK(unlink(f.c_str()) == 0);
    return 0;
}

int test_system_memory() {
    void* a = nullptr;
    void* b = nullptr;
    TS_CHECK(bfMalloc(&a, 4096, BF_SPACE_SYSTEM) == BF_STATUS_SUCCESS);
    TS_CHECK(bfMalloc(&b, 4096, BF_SPACE_SYSTEM) == BF_STATUS_SUCCESS);
    std::memset(a, 0x5A, 4096);
    TS_CHECK(bfMemcpy(b, BF_SPACE_SYSTEM, a, BF_SPACE_SYSTEM, 4096) ==
             BF_STATUS_SUCCESS);
    TS_CHECK(std::memcmp(a, b, 4096) == 0);
    TS_CHECK(bfMemset(a, BF_SPACE_SYSTEM, 0, 4096) == BF_STATUS_SUCCESS);
    TS_CHECK(((const unsigned char*)a)[0] == 0 &&
             ((const unsigned char*)a)[4095] == 0);
    bfFree(a, BF_SPACE_SYSTEM);
    bfFree(b, BF_SPACE_SYSTEM);
    return 0;
}

int test_ring_roundtrip() {
    const BFsize gulp = 64;
    BFring ring = nullptr;
    TS_CHECK(bfRingCreate(&ring, "bfamd_ts_ring", BF_SPACE_SYSTEM) ==
             BF_STATUS_SUCCESS);
    TS_CHECK(bfRingResize(ring, gulp, 4 * gulp, 1) == BF_STATUS_SUCCESS);
    TS_CHECK(bfRingBeginWriting(ring) == BF_STATUS_SUCCESS);

    BFwsequence ws = nullptr;
    const char hdr[] = "{\"ts\": 1}";
    TS_CHECK(bfRingSequenceBegin(&ws, ring, "ts_seq", 7,
                                 sizeof(hdr) - 1, hdr, 1, 0) ==
             BF_STATUS_SUCCESS);
    for (int i = 0; i < 3; ++i) {
        BFwspan span = nullptr;
        TS_CHECK(bfRingSpanReserve(&span, ring, gulp, 0) ==
                 BF_STATUS_SUCCESS);
        void* data = nullptr;
        TS_CHECK(bfRingSpanGetData((BFspan)span, &data) ==
                 BF_STATUS_SUCCESS);
        std::memset(data, i, gulp);
        TS_CHECK(bfRingSpanCommit(span, gulp) == BF_STATUS_SUCCESS);
    }
    TS_CHECK(bfRingSequenceEnd(ws, 0) == BF_STATUS_SUCCESS);
    TS_CHECK(bfRingEndWriting(ring) == BF_STATUS_SUCCESS);

    BFrsequence rs = nullptr;
    TS_CHECK(bfRingSequenceOpenEarliest(&rs, ring, 1) == BF_STATUS_SUCCESS);
    BFsequence_info sinfo;
    TS_CHECK(bfRingSequenceGetInfo((BFsequence)rs, &sinfo) ==
             BF_STATUS_SUCCESS);
    TS_CHECK(sinfo.time_tag == 7);
    TS_CHECK(sinfo.header_size == sizeof(hdr) - 1);
    TS_CHECK(std::memcmp(sinfo.header, hdr, sizeof(hdr) - 1) == 0);

    for (int i = 0; i < 3; ++i) {
        BFrspan span = nullptr;
        TS_CHECK(bfRingSpanAcquire(&span, rs, i * gulp, gulp) ==
                 BF_STATUS_SUCCESS);
        void* data = nullptr;
        BFsize size = 0;
        TS_CHECK(bfRingSpanGetData((BFspan)span, &data) ==
                 BF_STATUS_SUCCESS);
        TS_CHECK(bfRingSpanGetSize((BFspan)span, &size) ==
                 BF_STATUS_SUCCESS);
        TS_CHECK(size == gulp);
        TS_CHECK(((const unsigned char*)data)[0] == i &&
                 ((const unsigned char*)data)[gulp - 1] == i);
        TS_CHECK(bfRingSpanRelease(span) == BF_STATUS_SUCCESS);
    }
    BFrspan span = nullptr;
    TS_CHECK(bfRingSpanAcquire(&span, rs, 3 * gulp, gulp) ==
             BF_STATUS_END_OF_DATA);
    TS_CHECK(bfRingSequenceClose(rs) == BF_STATUS_SUCCESS);
    TS_CHECK(bfRingDestroy(ring) == BF_STATUS_SUCCESS);
    return 0;
}

}  // namespace

extern "C" int bfTestSuite() {
    int num_fails = 0;
    num_fails += test_this_process_exists();
    num_fails += test_make_then_remove_dir();
    num_fails += test_create_then_remove_file();
    num_fails += test_system_memory();
    num_fails += test_ring_roundtrip();
    return num_fails;
}
