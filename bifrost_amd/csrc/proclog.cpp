// bifrost_amd: per-block status files (read by tools like like_top.py).
// ABI: reference src/bifrost/proclog.h:40-43; files live under
// $BIFROST_PROCLOG_DIR or /dev/shm/bifrost_amd/<pid>/<name>.

#include <bifrost/proclog.h>

#include <sys/stat.h>
#include <unistd.h>

#include <cerrno>
#include <cstdio>
#include <cstring>
#include <string>

#include "status.hpp"

struct BFproclog_impl {
    std::string path;
};

namespace {

std::string proclog_root() {
    const char* env = std::getenv("BIFROST_PROCLOG_DIR");
    std::string base = env ? env : "/dev/shm/bifrost_amd";
    return base + "/" + std::to_string((long)getpid());
}

bool mkdirs(const std::string& path) {
    std::string cur;
    for (size_t i = 0; i < path.size(); ++i) {
        cur += path[i];
        if ((path[i] == '/' && i > 0) || i + 1 == path.size()) {
            if (mkdir(cur.c_str(), 0777) != 0 && errno != EEXIST &&
                errno != EISDIR) {
                return false;
            }
        }
    }
    return true;
}

}  // namespace

extern "C" {

BFstatus bfProcLogCreate(BFproclog* log_ptr, const char* name) {
    BF_ASSERT(log_ptr && name, BF_STATUS_INVALID_POINTER);
    BF_TRY_RETURN({
        std::string dir = proclog_root();
        std::string full = dir + "/" + name;
        // name may contain '/' sub-paths; create parents
        std::string parent = full.substr(0, full.find_last_of('/'));
        if (!mkdirs(parent)) throw bfamd::StatusError(BF_STATUS_INTERNAL_ERROR);
        auto* impl = new BFproclog_impl{full};
        *log_ptr = impl;
    });
}

BFstatus bfProcLogDestroy(BFproclog log) {
    BF_ASSERT(log, BF_STATUS_INVALID_HANDLE);
    std::remove(log->path.c_str());
    delete log;
    return BF_STATUS_SUCCESS;
}

BFstatus bfProcLogUpdate(BFproclog log, const char* str) {
    BF_ASSERT(log && str, BF_STATUS_INVALID_POINTER);
    FILE* f = std::fopen(log->path.c_str(), "w");
    BF_ASSERT(f, BF_STATUS_INTERNAL_ERROR);
    std::fputs(str, f);
    std::fclose(f);
    return BF_STATUS_SUCCESS;
}

}  // extern "C"
