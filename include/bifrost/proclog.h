/* bifrost_amd: per-block status files under BF_PROCLOG_DIR (tmpfs).
 * ABI identical to reference src/bifrost/proclog.h:37-43. */
#ifndef BFAMD_PROCLOG_H_
#define BFAMD_PROCLOG_H_

#include <bifrost/common.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct BFproclog_impl* BFproclog;

BFstatus bfProcLogCreate(BFproclog* log_ptr, const char* name);
BFstatus bfProcLogDestroy(BFproclog log);
BFstatus bfProcLogUpdate(BFproclog log, const char* str);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_PROCLOG_H_ */
