// HTTPS smoke: health + sync infer + async (epoll worker) infers over
// TLS with peer verification off (self-signed test cert).
// Usage: tls_smoke <host:port>
#include <chrono>
#include <condition_variable>
#include <iostream>
#include <memory>
#include <mutex>
#include <vector>

#include "client_amd/http_client.h"

using namespace client_amd;

int main(int argc, char** argv) {
  std::string url = argc > 1 ? argv[1] : "127.0.0.1:8443";
  HttpSslOptions ssl;
  ssl.verify_peer = false;
  ssl.verify_host = false;
  std::unique_ptr<InferenceServerHttpClient> client;
  Error err = InferenceServerHttpClient::Create(&client, url, false, true, ssl);
  if (!err.IsOk()) { std::cerr << err.Message() << std::endl; return 1; }
  bool live = false;
  err = client->IsServerLive(&live);
  if (!err.IsOk() || !live) {
    std::cerr << "live check failed: " << err.Message() << std::endl;
    return 1;
  }
  std::vector<int32_t> in0(16, 2), in1(16, 3);
  InferInput* input0;
  InferInput* input1;
  InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32");
  InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32");
  std::unique_ptr<InferInput> i0(input0), i1(input1);
  input0->AppendRaw((uint8_t*)in0.data(), 64);
  input1->AppendRaw((uint8_t*)in1.data(), 64);
  InferOptions options("simple");
  InferResult* result = nullptr;
  err = client->Infer(&result, options, {input0, input1});
  std::unique_ptr<InferResult> rp(result);
  if (!err.IsOk()) { std::cerr << err.Message() << std::endl; return 1; }
  const uint8_t* buf;
  size_t n;
  result->RawData("OUTPUT0", &buf, &n);
  if (((const int32_t*)buf)[0] != 5) {
    std::cerr << "wrong result over TLS" << std::endl;
    return 1;
  }
  // async path: several concurrent transfers, each with its own
  // non-blocking TLS handshake inside the epoll worker
  {
    // static: stack-reused std::mutex never runs pthread_mutex_destroy,
    // which poisons TSAN mutex shadow across test sections
    static std::mutex mu;
    static std::condition_variable cv;
    int done = 0;
    bool all_ok = true;
    const int kAsync = 6;
    for (int r = 0; r < kAsync; ++r) {
      err = client->AsyncInfer(
          [&](InferResult* res) {
            std::unique_ptr<InferResult> owned(res);
            const uint8_t* b;
            size_t bn;
            if (!owned->RequestStatus().IsOk() ||
                !owned->RawData("OUTPUT0", &b, &bn).IsOk() || bn != 64 ||
                ((const int32_t*)b)[7] != 5) {
              all_ok = false;
            }
            std::lock_guard<std::mutex> lock(mu);
            done++;
            cv.notify_all();
          },
          options, {input0, input1});
      if (!err.IsOk()) { std::cerr << err.Message() << std::endl; return 1; }
    }
    std::unique_lock<std::mutex> lock(mu);
    if (!cv.wait_for(lock, std::chrono::seconds(20),
                     [&] { return done == kAsync; })) {
      std::cerr << "async TLS transfers timed out (" << done << "/"
                << kAsync << ")" << std::endl;
      return 1;
    }
    if (!all_ok) {
      std::cerr << "async TLS result mismatch" << std::endl;
      return 1;
    }
  }
  std::cout << "PASS : https sync+async" << std::endl;
  return 0;
}
