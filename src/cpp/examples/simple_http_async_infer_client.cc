// Async HTTP inference via the epoll worker (reference:
// simple_http_async_infer_client.cc).
#include <condition_variable>
#include <iostream>
#include <memory>
#include <mutex>
#include <vector>

#include "client_amd/http_client.h"

namespace ca = client_amd;

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8000";
  for (int i = 1; i < argc - 1; ++i)
    if (std::string(argv[i]) == "-u") url = argv[i + 1];

  std::unique_ptr<ca::InferenceServerHttpClient> client;
  ca::InferenceServerHttpClient::Create(&client, url);

  std::vector<int32_t> in0(16), in1(16);
  for (int i = 0; i < 16; ++i) { in0[i] = i; in1[i] = 1; }
  ca::InferInput* input0;
  ca::InferInput* input1;
  ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32");
  ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  input0->AppendRaw((uint8_t*)in0.data(), 64);
  input1->AppendRaw((uint8_t*)in1.data(), 64);

  std::mutex mu;
  std::condition_variable cv;
  int done = 0;
  bool ok = true;
  ca::InferOptions options("simple");
  for (int r = 0; r < 4; ++r) {
    ca::Error err = client->AsyncInfer(
        [&](ca::InferResult* result) {
          std::unique_ptr<ca::InferResult> owned(result);
          if (!owned->RequestStatus().IsOk()) ok = false;
          std::lock_guard<std::mutex> lock(mu);
          done++;
          cv.notify_all();
        },
        options, {input0, input1});
    if (!err.IsOk()) { std::cerr << err.Message() << std::endl; return 1; }
  }
  std::unique_lock<std::mutex> lock(mu);
  cv.wait(lock, [&] { return done == 4; });
  if (!ok) { std::cerr << "async infer failed" << std::endl; return 1; }
  std::cout << "PASS : async infer" << std::endl;
  return 0;
}
