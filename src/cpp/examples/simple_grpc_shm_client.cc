// System shared-memory I/O over gRPC: no tensor bytes on the wire
// (reference: src/c++/examples/simple_grpc_shm_client.cc).
#include <cstring>

#include "client_amd/grpc_client.h"
#include "client_amd/shm_utils.h"
#include <iostream>
#include <memory>
#include <vector>

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
  client->UnregisterSystemSharedMemory();

  std::string key = "/simple_grpc_shm_example";
  int fd;
  FAIL_IF_ERR(ca::CreateSharedMemoryRegion(key, 256, &fd), "create region");
  void* base;
  FAIL_IF_ERR(ca::MapSharedMemory(fd, 0, 256, &base), "map region");
  int32_t* data = (int32_t*)base;
  for (int i = 0; i < 16; ++i) { data[i] = i; data[16 + i] = 1; }
  FAIL_IF_ERR(client->RegisterSystemSharedMemory("example_io", key, 256),
              "register");

  ca::InferInput* input0;
  ca::InferInput* input1;
  FAIL_IF_ERR(ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32"),
              "INPUT0");
  FAIL_IF_ERR(ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32"),
              "INPUT1");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  FAIL_IF_ERR(input0->SetSharedMemory("example_io", 64, 0), "shm 0");
  FAIL_IF_ERR(input1->SetSharedMemory("example_io", 64, 64), "shm 1");
  ca::InferRequestedOutput* output0;
  FAIL_IF_ERR(ca::InferRequestedOutput::Create(&output0, "OUTPUT0"), "out");
  std::unique_ptr<ca::InferRequestedOutput> op(output0);
  FAIL_IF_ERR(output0->SetSharedMemory("example_io", 64, 128), "out shm");

  ca::InferOptions options("simple");
  ca::InferResult* result;
  FAIL_IF_ERR(client->Infer(&result, options, {input0, input1}, {output0}),
              "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  const int32_t* sum = (const int32_t*)((const char*)base + 128);
  for (int i = 0; i < 16; ++i) {
    if (sum[i] != i + 1) { std::cerr << "mismatch" << std::endl; return 1; }
  }
  client->UnregisterSystemSharedMemory("example_io");
  ca::UnmapSharedMemory(base, 256);
  ca::CloseSharedMemory(fd);
  ca::UnlinkSharedMemoryRegion(key);
  std::cout << "PASS : grpc system shm" << std::endl;
  return 0;
}
