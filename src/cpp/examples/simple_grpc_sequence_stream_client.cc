// Stateful sequences over the bi-di stream (reference:
// simple_grpc_sequence_stream_infer_client.cc).
#include <chrono>
#include <condition_variable>
#include <iostream>
#include <memory>
#include <mutex>
#include <vector>

#include "client_amd/grpc_client.h"

namespace ca = client_amd;

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8001";
  for (int i = 1; i < argc - 1; ++i)
    if (std::string(argv[i]) == "-u") url = argv[i + 1];
  std::unique_ptr<ca::InferenceServerGrpcClient> client;
  ca::InferenceServerGrpcClient::Create(&client, url);

  std::mutex mu;
  std::condition_variable cv;
  std::vector<int32_t> seen;
  ca::Error start_err = client->StartStream([&](ca::InferResult* res) {
    std::unique_ptr<ca::InferResult> owned(res);
    const uint8_t* b;
    size_t n;
    if (owned->RequestStatus().IsOk() &&
        owned->RawData("OUTPUT", &b, &n).IsOk() && n >= 4) {
      std::lock_guard<std::mutex> lock(mu);
      seen.push_back(*(const int32_t*)b);
      cv.notify_all();
    }
  });
  if (!start_err.IsOk()) { std::cerr << start_err.Message() << std::endl; return 1; }

  int32_t values[4] = {11, 7, 5, 3};
  for (int i = 0; i < 4; ++i) {
    ca::InferInput* in;
    ca::InferInput::Create(&in, "INPUT", {1}, "INT32");
    std::unique_ptr<ca::InferInput> inp(in);
    in->AppendRaw((uint8_t*)&values[i], 4);
    ca::InferOptions opt("sequence_accumulate");
    opt.sequence_id_ = 1007;
    opt.sequence_start_ = (i == 0);
    opt.sequence_end_ = (i == 3);
    ca::Error err = client->AsyncStreamInfer(opt, {in});
    if (!err.IsOk()) { std::cerr << err.Message() << std::endl; return 1; }
  }
  {
    std::unique_lock<std::mutex> lock(mu);
    if (!cv.wait_for(lock, std::chrono::seconds(30),
                     [&] { return seen.size() == 4; })) {
      std::cerr << "timed out" << std::endl;
      return 1;
    }
  }
  client->StopStream();
  int32_t total = 0;
  for (int i = 0; i < 4; ++i) {
    total += values[i];
    if (seen[i] != total) { std::cerr << "wrong sum" << std::endl; return 1; }
  }
  std::cout << "PASS : sequence stream" << std::endl;
  return 0;
}
