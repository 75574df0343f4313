// BYTES/string inference (reference: simple_http_string_infer_client.cc).
#include <iostream>
#include <memory>
#include <vector>

#include "client_amd/http_client.h"

namespace ca = client_amd;

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8000";
  for (int i = 1; i < argc - 1; ++i)
    if (std::string(argv[i]) == "-u") url = argv[i + 1];
  std::unique_ptr<ca::InferenceServerHttpClient> client;
  ca::InferenceServerHttpClient::Create(&client, url);

  std::vector<std::string> s0, s1;
  for (int i = 0; i < 16; ++i) {
    s0.push_back(std::to_string(i));
    s1.push_back("1");
  }
  ca::InferInput* input0;
  ca::InferInput* input1;
  ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "BYTES");
  ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "BYTES");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  input0->AppendFromString(s0);
  input1->AppendFromString(s1);
  ca::InferOptions options("simple_string");
  ca::InferResult* result = nullptr;
  ca::Error err = client->Infer(&result, options, {input0, input1});
  std::unique_ptr<ca::InferResult> rp(result);
  if (!err.IsOk()) { std::cerr << err.Message() << std::endl; return 1; }
  std::vector<std::string> out;
  result->StringData("OUTPUT0", &out);
  for (int i = 0; i < 16; ++i) {
    if (out[i] != std::to_string(i + 1)) {
      std::cerr << "wrong result" << std::endl;
      return 1;
    }
  }
  std::cout << "PASS : string infer" << std::endl;
  return 0;
}
