/* bifrost_amd: stream/device control (HIP).  Symbol names keep the
 * reference's "cuda"-era spelling (src/bifrost/cuda.h:36-46) — on this
 * backend a "stream" is a hipStream_t and a device id is a HIP device.
 * Streams are per-host-thread (thread_local), set via bfStreamSet; every
 * op enqueues async on the calling thread's stream.
 */
#ifndef BFAMD_CUDA_H_
#define BFAMD_CUDA_H_

#include <bifrost/common.h>

#ifdef __cplusplus
extern "C" {
#endif

BFstatus bfStreamGet(void* stream);
BFstatus bfStreamSet(void const* stream);
BFstatus bfStreamSynchronize(void);
BFstatus bfDeviceGet(int* device);
BFstatus bfDeviceSet(int device);
BFstatus bfDeviceSetById(const char* pci_bus_id);
BFstatus bfDevicesSetNoSpinCPU(void);

#ifdef __cplusplus
}
#endif
#endif /* BFAMD_CUDA_H_ */
