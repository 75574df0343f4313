// Server health + metadata over HTTP (JSON)
// (reference: src/c++/examples/simple_http_health_metadata.cc).
#include "client_amd/http_client.h"
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
  std::string url = "127.0.0.1:8000";
  for (int i = 1; i < argc - 1; ++i)
    if (std::string(argv[i]) == "-u") url = argv[i + 1];

  std::unique_ptr<ca::InferenceServerHttpClient> client;
  FAIL_IF_ERR(ca::InferenceServerHttpClient::Create(&client, url), "create");
  bool live = false, ready = false, model_ready = false;
  FAIL_IF_ERR(client->IsServerLive(&live), "live");
  FAIL_IF_ERR(client->IsServerReady(&ready), "ready");
  FAIL_IF_ERR(client->IsModelReady(&model_ready, "simple"), "model ready");
  if (!live || !ready || !model_ready) {
    std::cerr << "server/model not ready" << std::endl;
    return 1;
  }
  std::string meta, config;
  FAIL_IF_ERR(client->ServerMetadata(&meta), "server metadata");
  FAIL_IF_ERR(client->ModelConfig(&config, "simple"), "model config");
  std::cout << "server metadata: " << meta << std::endl;
  std::cout << "PASS : health+metadata" << std::endl;
  return 0;
}
