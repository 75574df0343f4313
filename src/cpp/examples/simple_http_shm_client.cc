// System shared-memory I/O — no tensor bytes on the wire
// (reference: simple_http_shm_client.cc).
#include <cstring>
#include <iostream>
#include <memory>
#include <vector>

#include "client_amd/http_client.h"
#include "client_amd/shm_utils.h"

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

  client->UnregisterSystemSharedMemory();
  int fd;
  void* base;
  FAIL(ca::CreateSharedMemoryRegion("/simple_cc_shm", 256, &fd), "create");
  FAIL(ca::MapSharedMemory(fd, 0, 256, &base), "map");
  int32_t* data = (int32_t*)base;
  for (int i = 0; i < 16; ++i) { data[i] = i; data[16 + i] = 1; }
  FAIL(client->RegisterSystemSharedMemory("io", "/simple_cc_shm", 256),
       "register");

  ca::InferInput* input0;
  ca::InferInput* input1;
  ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32");
  ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  input0->SetSharedMemory("io", 64, 0);
  input1->SetSharedMemory("io", 64, 64);
  ca::InferRequestedOutput* out0;
  ca::InferRequestedOutput::Create(&out0, "OUTPUT0");
  std::unique_ptr<ca::InferRequestedOutput> o0(out0);
  out0->SetSharedMemory("io", 64, 128);

  ca::InferOptions options("simple");
  ca::InferResult* result = nullptr;
  FAIL(client->Infer(&result, options, {input0, input1}, {out0}), "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  for (int i = 0; i < 16; ++i) {
    if (data[32 + i] != data[i] + data[16 + i]) {
      std::cerr << "wrong shm result" << std::endl;
      return 1;
    }
  }
  client->UnregisterSystemSharedMemory("io");
  ca::UnmapSharedMemory(base, 256);
  ca::CloseSharedMemory(fd);
  ca::UnlinkSharedMemoryRegion("/simple_cc_shm");
  std::cout << "PASS : system shared memory" << std::endl;
  return 0;
}
