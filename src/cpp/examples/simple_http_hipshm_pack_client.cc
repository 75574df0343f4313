// The full MI355X data plane from C++: fp32 host data -> device scratch
// -> CDNA4 pack kernel (wire-exact fp32->bf16) into a HIP-IPC region ->
// BF16 inference with only region references on the wire -> unpack
// kernel on the output. Build with hipcc and kernels.hip (see
// include/client_amd/kernels.h). Requires a GPU + out-of-process server
// serving identity_bf16.
#include <cmath>
#include <cstring>
#include <iostream>
#include <memory>
#include <vector>

#include <hip/hip_runtime_api.h>

#include "client_amd/hip_shm.h"
#include "client_amd/http_client.h"
#include "client_amd/kernels.h"

namespace ca = client_amd;

#define FAIL(err, msg)                                             \
  {                                                                \
    ca::Error e = (err);                                           \
    if (!e.IsOk()) {                                               \
      std::cerr << msg << ": " << e.Message() << std::endl;        \
      return 1;                                                    \
    }                                                              \
  }
#define HIP_FAIL(expr, msg)                                        \
  {                                                                \
    hipError_t e = (expr);                                         \
    if (e != hipSuccess) {                                         \
      std::cerr << msg << ": " << hipGetErrorString(e) << std::endl; \
      return 1;                                                    \
    }                                                              \
  }

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8000";
  for (int i = 1; i < argc - 1; ++i)
    if (std::string(argv[i]) == "-u") url = argv[i + 1];
  std::unique_ptr<ca::InferenceServerHttpClient> client;
  ca::InferenceServerHttpClient::Create(&client, url);
  client->UnregisterCudaSharedMemory();

  const long n = 4096;
  std::vector<float> host(n);
  for (long i = 0; i < n; ++i) host[i] = 0.37f * (float)i - 100.f;

  // fp32 staging buffer + bf16 in/out regions, all in HBM
  void* staging;
  void* in_region;
  void* out_region;
  HIP_FAIL(hipMalloc(&staging, n * 4), "malloc staging");
  FAIL(ca::CreateHipSharedMemoryRegion(&in_region, n * 2, 0), "in region");
  FAIL(ca::CreateHipSharedMemoryRegion(&out_region, n * 2, 0), "out region");
  HIP_FAIL(hipMemcpy(staging, host.data(), n * 4, hipMemcpyHostToDevice),
           "h2d");
  // CDNA4 pack kernel: fp32 -> wire-exact bf16 straight into the region
  HIP_FAIL(ca_cast_fp32_bf16(staging, in_region, n, 0), "pack kernel");
  HIP_FAIL(hipStreamSynchronize(0), "sync");

  std::string in_handle, out_handle;
  FAIL(ca::GetHipSharedMemoryRegionHandle(&in_handle, in_region), "handle");
  FAIL(ca::GetHipSharedMemoryRegionHandle(&out_handle, out_region), "handle");
  FAIL(client->RegisterCudaSharedMemory("pack_in", in_handle, 0, n * 2),
       "register");
  FAIL(client->RegisterCudaSharedMemory("pack_out", out_handle, 0, n * 2),
       "register");

  ca::InferInput* input;
  ca::InferInput::Create(&input, "INPUT0", {n}, "BF16");
  std::unique_ptr<ca::InferInput> ip(input);
  input->SetSharedMemory("pack_in", n * 2);
  ca::InferRequestedOutput* output;
  ca::InferRequestedOutput::Create(&output, "OUTPUT0");
  std::unique_ptr<ca::InferRequestedOutput> op(output);
  output->SetSharedMemory("pack_out", n * 2);

  ca::InferOptions options("identity_bf16");
  ca::InferResult* result = nullptr;
  FAIL(client->Infer(&result, options, {input}, {output}), "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  FAIL(result->RequestStatus(), "status");

  // unpack kernel: bf16 region -> fp32, then verify truncation semantics
  HIP_FAIL(ca_cast_bf16_fp32(out_region, staging, n, 0), "unpack kernel");
  HIP_FAIL(hipStreamSynchronize(0), "sync");
  std::vector<float> out(n);
  HIP_FAIL(hipMemcpy(out.data(), staging, n * 4, hipMemcpyDeviceToHost),
           "d2h");
  for (long i = 0; i < n; ++i) {
    uint32_t bits;
    memcpy(&bits, &host[i], 4);
    bits &= 0xFFFF0000u;  // wire-exact truncation
    float expect;
    memcpy(&expect, &bits, 4);
    if (out[i] != expect) {
      std::cerr << "mismatch at " << i << ": " << out[i] << " != " << expect
                << std::endl;
      return 1;
    }
  }
  client->UnregisterCudaSharedMemory();
  ca::DestroyHipSharedMemoryRegion(in_region);
  ca::DestroyHipSharedMemoryRegion(out_region);
  HIP_FAIL(hipFree(staging), "free");
  std::cout << "PASS : hipshm pack kernel round trip" << std::endl;
  return 0;
}
