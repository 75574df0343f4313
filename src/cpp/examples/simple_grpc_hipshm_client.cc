// HIP-IPC shared memory over gRPC — tensors stay in HBM3E; the 64-byte
// hipIpcMemHandle_t rides raw in the proto. Requires 1 GPU and an
// out-of-process server (reference: simple_grpc_cudashm_client.cc —
// cuda_shared_memory IS hip_shared_memory on this stack).
#include <cstring>
#include <iostream>
#include <memory>
#include <vector>

#include "client_amd/grpc_client.h"
#include "client_amd/hip_shm.h"

namespace ca = client_amd;

#define FAIL_IF_ERR(X, MSG)                                      \
  {                                                              \
    ca::Error err = (X);                                         \
    if (!err.IsOk()) {                                           \
      std::cerr << "error: " << (MSG) << ": " << err.Message()   \
                << std::endl;                                    \
      exit(1);                                                   \
    }                                                            \
  }

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8001";
  for (int i = 1; i < argc - 1; ++i)
    if (std::string(argv[i]) == "-u") url = argv[i + 1];

  std::unique_ptr<ca::InferenceServerGrpcClient> client;
  FAIL_IF_ERR(ca::InferenceServerGrpcClient::Create(&client, url), "create");
  client->UnregisterCudaSharedMemory();

  void* in_ptr;
  void* out_ptr;
  FAIL_IF_ERR(ca::CreateHipSharedMemoryRegion(&in_ptr, 128, 0), "alloc in");
  FAIL_IF_ERR(ca::CreateHipSharedMemoryRegion(&out_ptr, 128, 0), "alloc out");
  std::string in_handle, out_handle;
  FAIL_IF_ERR(ca::GetHipSharedMemoryRegionHandle(&in_handle, in_ptr),
              "in handle");
  FAIL_IF_ERR(ca::GetHipSharedMemoryRegionHandle(&out_handle, out_ptr),
              "out handle");
  int32_t host[32];
  for (int i = 0; i < 16; ++i) { host[i] = i; host[16 + i] = 1; }
  FAIL_IF_ERR(ca::HipSharedMemoryRegionSet(in_ptr, 0, 128, host), "h2d");
  FAIL_IF_ERR(client->RegisterCudaSharedMemory("in_region", in_handle, 0, 128),
              "register in");
  FAIL_IF_ERR(
      client->RegisterCudaSharedMemory("out_region", out_handle, 0, 128),
      "register out");

  ca::InferInput* input0;
  ca::InferInput* input1;
  FAIL_IF_ERR(ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32"),
              "INPUT0");
  FAIL_IF_ERR(ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32"),
              "INPUT1");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  FAIL_IF_ERR(input0->SetSharedMemory("in_region", 64, 0), "shm 0");
  FAIL_IF_ERR(input1->SetSharedMemory("in_region", 64, 64), "shm 1");
  ca::InferRequestedOutput* output0;
  ca::InferRequestedOutput* output1;
  FAIL_IF_ERR(ca::InferRequestedOutput::Create(&output0, "OUTPUT0"), "o0");
  FAIL_IF_ERR(ca::InferRequestedOutput::Create(&output1, "OUTPUT1"), "o1");
  std::unique_ptr<ca::InferRequestedOutput> op0(output0), op1(output1);
  FAIL_IF_ERR(output0->SetSharedMemory("out_region", 64, 0), "o0 shm");
  FAIL_IF_ERR(output1->SetSharedMemory("out_region", 64, 64), "o1 shm");

  ca::InferOptions options("simple");
  ca::InferResult* result;
  FAIL_IF_ERR(
      client->Infer(&result, options, {input0, input1}, {output0, output1}),
      "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  int32_t out[32];
  FAIL_IF_ERR(ca::HipSharedMemoryRegionGet(out_ptr, 0, 128, out), "d2h");
  for (int i = 0; i < 16; ++i) {
    if (out[i] != i + 1 || out[16 + i] != i - 1) {
      std::cerr << "mismatch at " << i << std::endl;
      return 1;
    }
  }
  client->UnregisterCudaSharedMemory();
  ca::DestroyHipSharedMemoryRegion(in_ptr);
  ca::DestroyHipSharedMemoryRegion(out_ptr);
  std::cout << "PASS : grpc HIP shm" << std::endl;
  return 0;
}
