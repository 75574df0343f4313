// Callback-based async gRPC inference
// (reference: src/c++/examples/simple_grpc_async_infer_client.cc).
#include <condition_variable>
#include <mutex>

#include "client_amd/grpc_client.h"
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

  std::vector<int32_t> in0(16), in1(16);
  for (int i = 0; i < 16; ++i) { in0[i] = i; in1[i] = 1; }
  ca::InferInput* input0;
  ca::InferInput* input1;
  FAIL_IF_ERR(ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32"),
              "INPUT0");
  FAIL_IF_ERR(ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32"),
              "INPUT1");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  FAIL_IF_ERR(input0->AppendRaw((uint8_t*)in0.data(), 64), "set 0");
  FAIL_IF_ERR(input1->AppendRaw((uint8_t*)in1.data(), 64), "set 1");

  std::mutex mu;
  std::condition_variable cv;
  bool done = false, ok = false;
  ca::InferOptions options("simple");
  FAIL_IF_ERR(client->AsyncInfer(
      [&](ca::InferResult* result) {
        std::unique_ptr<ca::InferResult> rp(result);
        const uint8_t* out0;
        size_t n0;
        ok = result->RequestStatus().IsOk() &&
             result->RawData("OUTPUT0", &out0, &n0).IsOk() && n0 == 64 &&
             ((const int32_t*)out0)[5] == 6;
        std::lock_guard<std::mutex> lock(mu);
        done = true;
        cv.notify_all();
      },
      options, {input0, input1}), "async infer");
  std::unique_lock<std::mutex> lock(mu);
  cv.wait_for(lock, std::chrono::seconds(30), [&] { return done; });
  if (!ok) { std::cerr << "async result wrong" << std::endl; return 1; }
  std::cout << "PASS : async infer" << std::endl;
  return 0;
}
