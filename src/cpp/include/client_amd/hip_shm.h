// HIP-IPC shared-memory helpers for the C++ client — the MI355X data
// plane (replaces the reference's CUDA shared-memory example utilities,
// simple_http_cudashm_client.cc:107-114; the only GPU surface of the
// C++ library, cf. reference ipc.h:28-32 which stubs the handle type
// when built CPU-only — same pattern here with hipIpcMemHandle_t).
#pragma once

#include <string>

#include "client_amd/common.h"

#ifdef TRITON_ENABLE_HIP
#include <hip/hip_runtime_api.h>
#else
// CPU-only build: keep signatures compiling (reference ipc.h pattern)
struct hipIpcMemHandle_t_stub {
  char reserved[64];
};
using hipIpcMemHandle_t = hipIpcMemHandle_t_stub;
#endif

namespace client_amd {

// hipSetDevice + hipMalloc; returns the device base pointer.
Error CreateHipSharedMemoryRegion(
    void** base_addr, size_t byte_size, int device_id = 0);

// 64-byte hipIpcMemHandle_t for the region (raw bytes; base64 it for
// the HTTP register endpoint).
Error GetHipSharedMemoryRegionHandle(std::string* raw_handle, void* base_addr);

// hipMemcpy host -> region[offset]
Error HipSharedMemoryRegionSet(
    void* base_addr, size_t offset, size_t byte_size, const void* src);

// hipMemcpy region[offset] -> host
Error HipSharedMemoryRegionGet(
    void* base_addr, size_t offset, size_t byte_size, void* dst);

// hipFree
Error DestroyHipSharedMemoryRegion(void* base_addr);

// aliases keeping reference example spelling (cudashm examples)
inline Error CreateCudaSharedMemoryRegion(
    void** base_addr, size_t byte_size, int device_id = 0) {
  return CreateHipSharedMemoryRegion(base_addr, byte_size, device_id);
}

}  // namespace client_amd
