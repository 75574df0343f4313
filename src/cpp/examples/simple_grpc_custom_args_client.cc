// Custom request parameters on the wire (InferOptions
// request_parameters_; reference: simple_grpc_custom_args_client.cc).
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

  std::vector<int32_t> in0(16, 2), in1(16, 3);
  ca::InferInput* input0;
  ca::InferInput* input1;
  FAIL_IF_ERR(ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32"),
              "INPUT0");
  FAIL_IF_ERR(ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32"),
              "INPUT1");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  FAIL_IF_ERR(input0->AppendRaw((uint8_t*)in0.data(), 64), "set 0");
  FAIL_IF_ERR(input1->AppendRaw((uint8_t*)in1.data(), 64), "set 1");

  ca::InferOptions options("simple");
  options.request_id_ = "custom-args-1";
  options.request_parameters_["my_key"] = "my_value";
  options.request_parameters_["priority_hint"] = "7";
  ca::InferResult* result;
  FAIL_IF_ERR(client->Infer(&result, options, {input0, input1}), "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  std::string id;
  FAIL_IF_ERR(result->Id(&id), "id");
  if (id != "custom-args-1") { std::cerr << "id mismatch" << std::endl; return 1; }
  std::cout << "PASS : custom args" << std::endl;
  return 0;
}
