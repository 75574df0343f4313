// Memory-growth watch (reference src/c++/tests/memory_leak_test.cc):
// repeated sync + async inference loops on both transports (+ system
// shm churn) while watching VmRSS; growth beyond a budget after
// warm-up fails the test. The ASAN/LSAN pass (docs/SANITIZERS.md)
// catches literal leaks; this guards steady-state RSS drift — the same
// thing the reference runs under valgrind/RSS watch.
// Usage: memory_leak_test <http_host:port> <grpc_host:port> [iters]
#include <cstdio>
#include <cstring>
#include <fstream>
#include <iostream>
#include <memory>
#include <sstream>
#include <string>
#include <vector>

#include "client_amd/grpc_client.h"
#include "client_amd/http_client.h"
#include "client_amd/shm_utils.h"

using namespace client_amd;

#define CHECK_OK(expr)                                              \
  do {                                                              \
    Error e = (expr);                                               \
    if (!e.IsOk()) {                                                \
      std::cerr << "FAILED at " << __LINE__ << ": " << e.Message()  \
                << std::endl;                                       \
      return 1;                                                     \
    }                                                               \
  } while (0)

static long RssKb() {
  std::ifstream f("/proc/self/status");
  std::string line;
  while (std::getline(f, line)) {
    if (line.rfind("VmRSS:", 0) == 0) {
      return atol(line.c_str() + 6);
    }
  }
  return -1;
}

int main(int argc, char** argv) {
  std::string http_url = argc > 1 ? argv[1] : "127.0.0.1:8000";
  std::string grpc_url = argc > 2 ? argv[2] : "127.0.0.1:8001";
  int iters = argc > 3 ? atoi(argv[3]) : 400;

  std::unique_ptr<InferenceServerHttpClient> http;
  std::unique_ptr<InferenceServerGrpcClient> grpc;
  CHECK_OK(InferenceServerHttpClient::Create(&http, http_url));
  CHECK_OK(InferenceServerGrpcClient::Create(&grpc, grpc_url));

  std::vector<int32_t> in0(16, 2), in1(16, 3);
  InferInput* input0;
  InferInput* input1;
  CHECK_OK(InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32"));
  CHECK_OK(InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32"));
  std::unique_ptr<InferInput> i0(input0), i1(input1);
  CHECK_OK(input0->AppendRaw((uint8_t*)in0.data(), 64));
  CHECK_OK(input1->AppendRaw((uint8_t*)in1.data(), 64));
  InferOptions options("simple");

  auto loop = [&](int n) -> Error {
    for (int i = 0; i < n; ++i) {
      InferResult* r = nullptr;
      RETURN_IF_ERROR(http->Infer(&r, options, {input0, input1}));
      delete r;
      r = nullptr;
      RETURN_IF_ERROR(grpc->Infer(&r, options, {input0, input1}));
      delete r;
      if (i % 16 == 0) {
        // shm register/unregister churn
        std::string key = "/mlt_shm";
        int fd;
        RETURN_IF_ERROR(CreateSharedMemoryRegion(key, 256, &fd));
        void* base;
        RETURN_IF_ERROR(MapSharedMemory(fd, 0, 256, &base));
        RETURN_IF_ERROR(grpc->RegisterSystemSharedMemory("mlt", key, 256));
        RETURN_IF_ERROR(grpc->UnregisterSystemSharedMemory("mlt"));
        RETURN_IF_ERROR(UnmapSharedMemory(base, 256));
        RETURN_IF_ERROR(CloseSharedMemory(fd));
        RETURN_IF_ERROR(UnlinkSharedMemoryRegion(key));
      }
    }
    return Error::Success;
  };

  // warm-up: allocator pools, connection buffers, HPACK tables settle
  CHECK_OK(loop(iters / 4));
  long before = RssKb();
  CHECK_OK(loop(iters));
  long after = RssKb();
  long growth = after - before;
  printf("RSS before=%ldKB after=%ldKB growth=%ldKB over %d iters\n",
         before, after, growth, iters);
  // budget: steady-state growth must stay under 4 MB (reference
  // memory_leak_test uses the same order of slack for allocator noise)
  if (growth > 4096) {
    fprintf(stderr, "FAILED: RSS grew %ldKB\n", growth);
    return 1;
  }
  printf("memory_leak_test: ALL PASSED\n");
  return 0;
}
