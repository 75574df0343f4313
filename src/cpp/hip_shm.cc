#include "client_amd/hip_shm.h"

#ifdef TRITON_ENABLE_HIP

#include <hip/hip_runtime_api.h>

namespace client_amd {

namespace {
Error HipErr(const char* what, hipError_t e) {
  return Error(std::string(what) + " failed: " + hipGetErrorString(e));
}
}  // namespace

Error CreateHipSharedMemoryRegion(
    void** base_addr, size_t byte_size, int device_id) {
  hipError_t e = hipSetDevice(device_id);
  if (e != hipSuccess) return HipErr("hipSetDevice", e);
  e = hipMalloc(base_addr, byte_size);
  if (e != hipSuccess) return HipErr("hipMalloc", e);
  return Error::Success;
}

Error GetHipSharedMemoryRegionHandle(
    std::string* raw_handle, void* base_addr) {
  hipIpcMemHandle_t handle;
  hipError_t e = hipIpcGetMemHandle(&handle, base_addr);
  if (e != hipSuccess) return HipErr("hipIpcGetMemHandle", e);
  raw_handle->assign(reinterpret_cast<const char*>(&handle), sizeof(handle));
  return Error::Success;
}

Error HipSharedMemoryRegionSet(
    void* base_addr, size_t offset, size_t byte_size, const void* src) {
  hipError_t e = hipMemcpy(
      static_cast<char*>(base_addr) + offset, src, byte_size,
      hipMemcpyHostToDevice);
  if (e != hipSuccess) return HipErr("hipMemcpy H2D", e);
  return Error::Success;
}

Error HipSharedMemoryRegionGet(
    void* base_addr, size_t offset, size_t byte_size, void* dst) {
  hipError_t e = hipMemcpy(
      dst, static_cast<char*>(base_addr) + offset, byte_size,
      hipMemcpyDeviceToHost);
  if (e != hipSuccess) return HipErr("hipMemcpy D2H", e);
  return Error::Success;
}

Error DestroyHipSharedMemoryRegion(void* base_addr) {
  hipError_t e = hipFree(base_addr);
  if (e != hipSuccess) return HipErr("hipFree", e);
  return Error::Success;
}

}  // namespace client_amd

#else  // CPU-only build

namespace client_amd {

static Error NoHip() {
  return Error("client_amd built without TRITON_ENABLE_HIP");
}

Error CreateHipSharedMemoryRegion(void**, size_t, int) { return NoHip(); }
Error GetHipSharedMemoryRegionHandle(std::string*, void*) { return NoHip(); }
Error HipSharedMemoryRegionSet(void*, size_t, size_t, const void*) {
  return NoHip();
}
Error HipSharedMemoryRegionGet(void*, size_t, size_t, void*) {
  return NoHip();
}
Error DestroyHipSharedMemoryRegion(void*) { return NoHip(); }

}  // namespace client_amd

#endif
