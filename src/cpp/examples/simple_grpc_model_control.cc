// Explicit model load/unload + repository index over gRPC
// (reference: src/c++/examples/simple_grpc_model_control.cc).
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
  FAIL_IF_ERR(client->UnloadModel("simple"), "unload");
  bool ready = true;
  FAIL_IF_ERR(client->IsModelReady(&ready, "simple"), "ready check");
  if (ready) { std::cerr << "still ready after unload" << std::endl; return 1; }
  FAIL_IF_ERR(client->LoadModel("simple"), "load");
  FAIL_IF_ERR(client->IsModelReady(&ready, "simple"), "ready check");
  if (!ready) { std::cerr << "not ready after load" << std::endl; return 1; }
  std::vector<ca::kserve::RepositoryIndexEntryPb> index;
  FAIL_IF_ERR(client->ModelRepositoryIndex(&index), "index");
  for (const auto& e : index)
    std::cout << e.name << " : " << e.state << std::endl;
  std::cout << "PASS : model control" << std::endl;
  return 0;
}
