// Decoupled model over the bi-di stream: one request, N streamed
// responses (reference: src/c++/examples/simple_grpc_custom_repeat.cc).
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

  std::mutex mu;
  std::condition_variable cv;
  std::vector<int32_t> seen;
  FAIL_IF_ERR(client->StartStream([&](ca::InferResult* result) {
    std::unique_ptr<ca::InferResult> rp(result);
    const uint8_t* out;
    size_t n;
    if (result->RequestStatus().IsOk() &&
        result->RawData("OUT", &out, &n).IsOk() && n >= 4) {
      std::lock_guard<std::mutex> lock(mu);
      seen.push_back(*(const int32_t*)out);
      cv.notify_all();
    }
  }), "start stream");

  const int kRepeat = 4;
  std::vector<int32_t> values = {10, 20, 30, 40};
  uint32_t delay = 0;
  ca::InferInput* in_vals;
  ca::InferInput* in_delay;
  FAIL_IF_ERR(ca::InferInput::Create(&in_vals, "IN", {kRepeat}, "INT32"),
              "IN");
  FAIL_IF_ERR(ca::InferInput::Create(&in_delay, "DELAY", {1}, "UINT32"),
              "DELAY");
  std::unique_ptr<ca::InferInput> iv(in_vals), id(in_delay);
  FAIL_IF_ERR(in_vals->AppendRaw((uint8_t*)values.data(), kRepeat * 4),
              "set IN");
  FAIL_IF_ERR(in_delay->AppendRaw((uint8_t*)&delay, 4), "set DELAY");
  ca::InferOptions options("repeat_int32");
  FAIL_IF_ERR(client->AsyncStreamInfer(options, {in_vals, in_delay}),
              "stream infer");
  {
    std::unique_lock<std::mutex> lock(mu);
    if (!cv.wait_for(lock, std::chrono::seconds(30),
                     [&] { return seen.size() >= (size_t)kRepeat; })) {
      std::cerr << "timed out (" << seen.size() << ")" << std::endl;
      return 1;
    }
  }
  FAIL_IF_ERR(client->StopStream(), "stop stream");
  for (int i = 0; i < kRepeat; ++i) {
    if (seen[i] != values[i]) { std::cerr << "mismatch" << std::endl; return 1; }
  }
  std::cout << "PASS : decoupled repeat" << std::endl;
  return 0;
}
