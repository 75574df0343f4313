// Stateful sequences with sync HTTP calls
// (reference: simple_http_sequence_sync_infer_client.cc).
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

  int32_t values[4] = {4, 2, 0, 1};
  int32_t total = 0;
  for (int i = 0; i < 4; ++i) {
    ca::InferInput* input;
    FAIL_IF_ERR(ca::InferInput::Create(&input, "INPUT", {1}, "INT32"),
                "INPUT");
    std::unique_ptr<ca::InferInput> ip(input);
    FAIL_IF_ERR(input->AppendRaw((uint8_t*)&values[i], 4), "set");
    ca::InferOptions options("sequence_accumulate");
    options.sequence_id_ = 98;
    options.sequence_start_ = (i == 0);
    options.sequence_end_ = (i == 3);
    ca::InferResult* result;
    FAIL_IF_ERR(client->Infer(&result, options, {input}), "infer");
    std::unique_ptr<ca::InferResult> rp(result);
    const uint8_t* out;
    size_t n;
    FAIL_IF_ERR(result->RawData("OUTPUT", &out, &n), "OUTPUT");
    total += values[i];
    if (*(const int32_t*)out != total) {
      std::cerr << "wrong accumulation" << std::endl;
      return 1;
    }
  }
  std::cout << "PASS : sequence sync" << std::endl;
  return 0;
}
