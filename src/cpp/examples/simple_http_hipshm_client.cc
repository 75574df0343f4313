// HIP-IPC shared-memory I/O: tensors stay in the MI355X's HBM3E
// (reference: simple_http_cudashm_client.cc:107-114 — on this stack the
// region is hipMalloc'd and the server opens it with
// hipIpcOpenMemHandle). Requires a GPU and an out-of-process server.
#include <cstring>
#include <iostream>
#include <memory>
#include <vector>

#include "client_amd/base64.h"
#include "client_amd/hip_shm.h"
#include "client_amd/http_client.h"

namespace ca = client_amd;

#define FAIL(err, msg)                                             \
  {                                                                \
    ca::Error e = (err);                                           \
    if (!e.IsOk()) {                                               \
      std::cerr << msg << ": " << e.Message() << std::endl;        \
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
  void* in_base;
  void* out_base;
  FAIL(ca::CreateHipSharedMemoryRegion(&in_base, 128, 0), "hipMalloc in");
  FAIL(ca::CreateHipSharedMemoryRegion(&out_base, 128, 0), "hipMalloc out");
  std::vector<int32_t> in0(16), in1(16);
  for (int i = 0; i < 16; ++i) { in0[i] = i; in1[i] = 1; }
  FAIL(ca::HipSharedMemoryRegionSet(in_base, 0, 64, in0.data()), "h2d 0");
  FAIL(ca::HipSharedMemoryRegionSet(in_base, 64, 64, in1.data()), "h2d 1");
  std::string in_handle, out_handle;
  FAIL(ca::GetHipSharedMemoryRegionHandle(&in_handle, in_base), "handle in");
  FAIL(ca::GetHipSharedMemoryRegionHandle(&out_handle, out_base),
       "handle out");
  FAIL(client->RegisterCudaSharedMemory("input_data", in_handle, 0, 128),
       "register in");
  FAIL(client->RegisterCudaSharedMemory("output_data", out_handle, 0, 128),
       "register out");

  ca::InferInput* input0;
  ca::InferInput* input1;
  ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32");
  ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  input0->SetSharedMemory("input_data", 64, 0);
  input1->SetSharedMemory("input_data", 64, 64);
  ca::InferRequestedOutput* out0;
  ca::InferRequestedOutput* out1;
  ca::InferRequestedOutput::Create(&out0, "OUTPUT0");
  ca::InferRequestedOutput::Create(&out1, "OUTPUT1");
  std::unique_ptr<ca::InferRequestedOutput> o0(out0), o1(out1);
  out0->SetSharedMemory("output_data", 64, 0);
  out1->SetSharedMemory("output_data", 64, 64);

  ca::InferOptions options("simple");
  ca::InferResult* result = nullptr;
  FAIL(client->Infer(&result, options, {input0, input1}, {out0, out1}),
       "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  std::vector<int32_t> sum(16), diff(16);
  FAIL(ca::HipSharedMemoryRegionGet(out_base, 0, 64, sum.data()), "d2h 0");
  FAIL(ca::HipSharedMemoryRegionGet(out_base, 64, 64, diff.data()), "d2h 1");
  for (int i = 0; i < 16; ++i) {
    if (sum[i] != in0[i] + in1[i] || diff[i] != in0[i] - in1[i]) {
      std::cerr << "wrong hipshm result" << std::endl;
      return 1;
    }
  }
  client->UnregisterCudaSharedMemory();
  ca::DestroyHipSharedMemoryRegion(in_base);
  ca::DestroyHipSharedMemoryRegion(out_base);
  std::cout << "PASS : hip shared memory" << std::endl;
  return 0;
}
