// Reuse InferInput/InferRequestedOutput across requests
// (reference: reuse_infer_objects_client.cc).
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

  ca::InferInput* input0;
  ca::InferInput* input1;
  ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32");
  ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32");
  std::unique_ptr<ca::InferInput> i0(input0), i1(input1);
  ca::InferRequestedOutput* out0;
  ca::InferRequestedOutput::Create(&out0, "OUTPUT0");
  std::unique_ptr<ca::InferRequestedOutput> o0(out0);
  ca::InferOptions options("simple");

  for (int trial = 0; trial < 3; ++trial) {
    std::vector<int32_t> in0(16, trial), in1(16, 1);
    input0->Reset();
    input1->Reset();
    input0->AppendRaw((uint8_t*)in0.data(), 64);
    input1->AppendRaw((uint8_t*)in1.data(), 64);
    ca::InferResult* result = nullptr;
    ca::Error err = client->Infer(&result, options, {input0, input1}, {out0});
    std::unique_ptr<ca::InferResult> rp(result);
    if (!err.IsOk()) { std::cerr << err.Message() << std::endl; return 1; }
    const uint8_t* buf;
    size_t n;
    result->RawData("OUTPUT0", &buf, &n);
    if (((const int32_t*)buf)[5] != trial + 1) {
      std::cerr << "wrong result" << std::endl;
      return 1;
    }
  }
  std::cout << "PASS : reuse infer objects" << std::endl;
  return 0;
}
