// BYTES/string tensors over gRPC: AppendFromString / StringData
// (reference: src/c++/examples/simple_grpc_string_infer_client.cc).
#include <string>

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

  std::vector<std::string> in0, in1;
  for (int i = 0; i < 16; ++i) {
    in0.push_back(std::to_string(i));
    in1.push_back("1");
  }
  ca::InferInput* input0;
  ca::InferInput* input1;
  FAIL_IF_ERR(ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "BYTES"),
              "INPUT0");
  FAIL_IF_ERR(ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "BYTES"),
              "INPUT1");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  FAIL_IF_ERR(input0->AppendFromString(in0), "set 0");
  FAIL_IF_ERR(input1->AppendFromString(in1), "set 1");

  ca::InferOptions options("simple_string");
  ca::InferResult* result;
  FAIL_IF_ERR(client->Infer(&result, options, {input0, input1}), "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  std::vector<std::string> out0;
  FAIL_IF_ERR(result->StringData("OUTPUT0", &out0), "OUTPUT0");
  for (int i = 0; i < 16; ++i) {
    if (std::stoi(out0[i]) != i + 1) {
      std::cerr << "mismatch at " << i << ": " << out0[i] << std::endl;
      return 1;
    }
  }
  std::cout << "PASS : string infer" << std::endl;
  return 0;
}
