// gRPC KeepAlive: h2 PING watchdog on a private channel
// (reference: src/c++/examples/simple_grpc_keepalive_client.cc).
#include <thread>

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

  ca::KeepAliveOptions keepalive;
  keepalive.keepalive_time_ms = 200;
  keepalive.keepalive_timeout_ms = 5000;
  keepalive.keepalive_permit_without_calls = true;
  keepalive.http2_max_pings_without_data = 0;
  std::unique_ptr<ca::InferenceServerGrpcClient> client;
  FAIL_IF_ERR(ca::InferenceServerGrpcClient::Create(
                  &client, url, false, false, ca::SslOptions(), keepalive,
                  /*use_cached_channel=*/false),
              "create");

  bool live = false;
  FAIL_IF_ERR(client->IsServerLive(&live), "live");
  std::this_thread::sleep_for(std::chrono::milliseconds(700));
  FAIL_IF_ERR(client->IsServerLive(&live), "live after pings");
  if (!live) { std::cerr << "not live" << std::endl; return 1; }
  std::cout << "PASS : keepalive" << std::endl;
  return 0;
}
