// Minimal sync HTTP inference example against the "simple" addsub
// model (reference: src/c++/examples/simple_http_infer_client.cc).
// Usage: simple_http_infer_client [-u host:port]
#include <cstring>
#include <iostream>
#include <memory>
#include <vector>

#include "client_amd/http_client.h"

namespace ca = client_amd;

#define FAIL_IF_ERR(X, MSG)                               \
  {                                                       \
    ca::Error err = (X);                                  \
    if (!err.IsOk()) {                                    \
      std::cerr << "error: " << (MSG) << ": "             \
                << err.Message() << std::endl;            \
      exit(1);                                            \
    }                                                     \
  }

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8000";
  for (int i = 1; i < argc - 1; ++i) {
    if (std::string(argv[i]) == "-u") url = argv[i + 1];
  }

  std::unique_ptr<ca::InferenceServerHttpClient> client;
  FAIL_IF_ERR(
      ca::InferenceServerHttpClient::Create(&client, url),
      "unable to create client");

  std::vector<int32_t> input0_data(16);
  std::vector<int32_t> input1_data(16);
  for (int i = 0; i < 16; ++i) {
    input0_data[i] = i;
    input1_data[i] = 1;
  }

  ca::InferInput* input0;
  ca::InferInput* input1;
  FAIL_IF_ERR(
      ca::InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32"),
      "unable to create INPUT0");
  FAIL_IF_ERR(
      ca::InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32"),
      "unable to create INPUT1");
  std::unique_ptr<ca::InferInput> input0_ptr(input0), input1_ptr(input1);
  FAIL_IF_ERR(
      input0->AppendRaw(
          reinterpret_cast<uint8_t*>(input0_data.data()),
          input0_data.size() * sizeof(int32_t)),
      "setting INPUT0 data");
  FAIL_IF_ERR(
      input1->AppendRaw(
          reinterpret_cast<uint8_t*>(input1_data.data()),
          input1_data.size() * sizeof(int32_t)),
      "setting INPUT1 data");

  ca::InferOptions options("simple");
  ca::InferResult* result;
  FAIL_IF_ERR(
      client->Infer(&result, options, {input0, input1}), "infer failed");
  std::unique_ptr<ca::InferResult> result_ptr(result);
  FAIL_IF_ERR(result->RequestStatus(), "request failed");

  const uint8_t* out0;
  const uint8_t* out1;
  size_t n0, n1;
  FAIL_IF_ERR(result->RawData("OUTPUT0", &out0, &n0), "OUTPUT0");
  FAIL_IF_ERR(result->RawData("OUTPUT1", &out1, &n1), "OUTPUT1");
  const int32_t* sum = reinterpret_cast<const int32_t*>(out0);
  const int32_t* diff = reinterpret_cast<const int32_t*>(out1);
  for (int i = 0; i < 16; ++i) {
    std::cout << input0_data[i] << " + " << input1_data[i] << " = " << sum[i]
              << "    " << input0_data[i] << " - " << input1_data[i] << " = "
              << diff[i] << std::endl;
    if (sum[i] != input0_data[i] + input1_data[i] ||
        diff[i] != input0_data[i] - input1_data[i]) {
      std::cerr << "error: incorrect result" << std::endl;
      return 1;
    }
  }
  std::cout << "PASS : infer" << std::endl;
  return 0;
}
