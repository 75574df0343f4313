// Explicit model load/unload over HTTP
// (reference: src/c++/examples/simple_http_model_control.cc).
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
  FAIL_IF_ERR(client->UnloadModel("simple"), "unload");
  bool ready = true;
  FAIL_IF_ERR(client->IsModelReady(&ready, "simple"), "ready check");
  if (ready) { std::cerr << "still ready after unload" << std::endl; return 1; }
  FAIL_IF_ERR(client->LoadModel("simple"), "load");
  FAIL_IF_ERR(client->IsModelReady(&ready, "simple"), "ready check");
  if (!ready) { std::cerr << "not ready after load" << std::endl; return 1; }
  std::string index;
  FAIL_IF_ERR(client->ModelRepositoryIndex(&index), "index");
  std::cout << index << std::endl;
  std::cout << "PASS : model control" << std::endl;
  return 0;
}
