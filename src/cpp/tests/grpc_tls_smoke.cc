// gRPC-over-TLS smoke: full peer verification of a self-signed root
// (args: host port root_cert.pem). Exercises the OpenSSL+ALPN-h2 path
// in h2.cc against a grpcio secure port.
#include <cstdio>
#include <cstdlib>
#include <fstream>
#include <memory>
#include <sstream>
#include <vector>

#include "client_amd/grpc_client.h"

using namespace client_amd;

#define CHECK_OK(err)                                       \
  do {                                                      \
    const Error& e_ = (err);                                \
    if (!e_.IsOk()) {                                       \
      fprintf(stderr, "FAIL: %s\n", e_.Message().c_str());  \
      return 1;                                             \
    }                                                       \
  } while (0)

static std::string ReadFile(const char* path) {
  std::ifstream f(path);
  std::stringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

int main(int argc, char** argv) {
  if (argc != 4 && argc != 6) {
    fprintf(stderr,
            "usage: %s host port root_cert.pem [client_key.pem "
            "client_cert.pem]\n",
            argv[0]);
    return 2;
  }
  SslOptions ssl;
  ssl.root_certificates = ReadFile(argv[3]);
  if (argc == 6) {  // mTLS
    ssl.private_key = ReadFile(argv[4]);
    ssl.certificate_chain = ReadFile(argv[5]);
  }
  std::unique_ptr<InferenceServerGrpcClient> client;
  CHECK_OK(InferenceServerGrpcClient::Create(
      &client, std::string(argv[1]) + ":" + argv[2], false, true, ssl));

  bool live = false;
  CHECK_OK(client->IsServerLive(&live));
  if (!live) {
    fprintf(stderr, "FAIL: server not live\n");
    return 1;
  }

  std::vector<int32_t> a(16, 2), b(16, 3);
  InferInput* in0;
  InferInput* in1;
  CHECK_OK(InferInput::Create(&in0, "INPUT0", {1, 16}, "INT32"));
  CHECK_OK(InferInput::Create(&in1, "INPUT1", {1, 16}, "INT32"));
  std::unique_ptr<InferInput> in0_owner(in0), in1_owner(in1);
  CHECK_OK(in0->AppendRaw((const uint8_t*)a.data(), a.size() * 4));
  CHECK_OK(in1->AppendRaw((const uint8_t*)b.data(), b.size() * 4));

  InferOptions options("simple");
  InferResult* result = nullptr;
  CHECK_OK(client->Infer(&result, options, {in0, in1}));
  std::unique_ptr<InferResult> result_owner(result);
  CHECK_OK(result->RequestStatus());

  const uint8_t* buf;
  size_t byte_size;
  CHECK_OK(result->RawData("OUTPUT0", &buf, &byte_size));
  if (byte_size != 64) {
    fprintf(stderr, "FAIL: OUTPUT0 byte_size %zu\n", byte_size);
    return 1;
  }
  const int32_t* sum = (const int32_t*)buf;
  for (int i = 0; i < 16; ++i) {
    if (sum[i] != 5) {
      fprintf(stderr, "FAIL: OUTPUT0[%d] = %d\n", i, sum[i]);
      return 1;
    }
  }
  printf("PASS\n");
  return 0;
}
