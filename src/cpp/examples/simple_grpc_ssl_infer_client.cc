// gRPC-over-TLS example: reference simple_grpc_infer_client with the
// SSL flags (-ssl, --root-certificates/--private-key/--certificate-chain
// taking PEM file paths, read into SslOptions as the reference examples
// do with ReadFile).
#include <fstream>
#include <iostream>
#include <memory>
#include <sstream>
#include <vector>

#include "client_amd/grpc_client.h"

using namespace client_amd;

static std::string ReadFile(const std::string& path) {
  std::ifstream f(path);
  std::stringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8001";
  SslOptions ssl;
  bool use_ssl = false;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "-u" && i + 1 < argc) url = argv[++i];
    else if (a == "-ssl") use_ssl = true;
    else if (a == "--root-certificates" && i + 1 < argc)
      ssl.root_certificates = ReadFile(argv[++i]);
    else if (a == "--private-key" && i + 1 < argc)
      ssl.private_key = ReadFile(argv[++i]);
    else if (a == "--certificate-chain" && i + 1 < argc)
      ssl.certificate_chain = ReadFile(argv[++i]);
  }

  std::unique_ptr<InferenceServerGrpcClient> client;
  Error err = InferenceServerGrpcClient::Create(
      &client, url, false, use_ssl, ssl);
  if (!err.IsOk()) { std::cerr << err << std::endl; return 1; }

  std::vector<int32_t> in0(16), in1(16);
  for (int i = 0; i < 16; ++i) { in0[i] = i; in1[i] = 1; }
  InferInput* input0;
  InferInput* input1;
  InferInput::Create(&input0, "INPUT0", {1, 16}, "INT32");
  InferInput::Create(&input1, "INPUT1", {1, 16}, "INT32");
  std::unique_ptr<InferInput> i0(input0), i1(input1);
  input0->AppendRaw((uint8_t*)in0.data(), 64);
  input1->AppendRaw((uint8_t*)in1.data(), 64);

  InferOptions options("simple");
  InferResult* result = nullptr;
  err = client->Infer(&result, options, {input0, input1});
  std::unique_ptr<InferResult> rp(result);
  if (!err.IsOk()) { std::cerr << err << std::endl; return 1; }
  const uint8_t* buf;
  size_t n;
  result->RawData("OUTPUT0", &buf, &n);
  const int32_t* sum = (const int32_t*)buf;
  for (int i = 0; i < 16; ++i) {
    std::cout << in0[i] << " + " << in1[i] << " = " << sum[i] << std::endl;
    if (sum[i] != in0[i] + in1[i]) { std::cerr << "MISMATCH\n"; return 1; }
  }
  std::cout << "PASS" << std::endl;
  return 0;
}
