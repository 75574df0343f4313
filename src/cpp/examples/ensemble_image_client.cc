// Ensemble pipeline client (reference: src/c++/examples/
// ensemble_image_client.cc, sans OpenCV): ships the RAW HWC uint8 image
// to the server's "ensemble_image" pipeline — the HIP preprocess kernel
// (bilinear resize + INCEPTION scaling + CHW pack) and the classifier
// both run server-side on the MI355X; the client never preprocesses.
// Usage: ensemble_image_client [-u host:port] [-c topk] [image.ppm]
#include <cstdint>
#include <fstream>
#include <iostream>
#include <memory>
#include <random>
#include <string>
#include <vector>

#include "client_amd/http_client.h"

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

static bool LoadPpm(const std::string& path, std::vector<uint8_t>* pix,
                    int* h, int* w) {
  std::ifstream f(path, std::ios::binary);
  if (!f) return false;
  std::string magic;
  int maxval = 0;
  f >> magic >> *w >> *h >> maxval;
  if (magic != "P6" || maxval != 255) return false;
  f.get();
  pix->resize((size_t)*h * *w * 3);
  f.read((char*)pix->data(), pix->size());
  return (bool)f;
}

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8000", image_path;
  int topk = 3;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "-u" && i + 1 < argc) url = argv[++i];
    else if (a == "-c" && i + 1 < argc) topk = atoi(argv[++i]);
    else if (a[0] != '-') image_path = a;
  }

  int ih = 480, iw = 640;
  std::vector<uint8_t> img;
  if (!image_path.empty()) {
    if (!LoadPpm(image_path, &img, &ih, &iw)) {
      std::cerr << "failed to read PPM " << image_path << std::endl;
      return 1;
    }
  } else {
    img.resize((size_t)ih * iw * 3);
    std::mt19937 rng(11);
    for (auto& b : img) b = (uint8_t)(rng() & 0xFF);
  }

  std::unique_ptr<ca::InferenceServerHttpClient> client;
  FAIL_IF_ERR(ca::InferenceServerHttpClient::Create(&client, url), "create");

  ca::InferInput* input;
  FAIL_IF_ERR(ca::InferInput::Create(&input, "IMAGE", {ih, iw, 3}, "UINT8"),
              "IMAGE");
  std::unique_ptr<ca::InferInput> ip(input);
  FAIL_IF_ERR(input->AppendRaw(img.data(), img.size()), "set");

  ca::InferRequestedOutput* output;
  FAIL_IF_ERR(ca::InferRequestedOutput::Create(&output, "OUTPUT0", topk),
              "OUTPUT0");
  std::unique_ptr<ca::InferRequestedOutput> op(output);

  ca::InferOptions options("ensemble_image");
  ca::InferResult* result;
  FAIL_IF_ERR(client->Infer(&result, options, {input}, {output}), "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  std::vector<std::string> classes;
  FAIL_IF_ERR(result->StringData("OUTPUT0", &classes), "classes");
  for (const auto& c : classes) std::cout << "    " << c << std::endl;
  if ((int)classes.size() != topk) {
    std::cerr << "expected " << topk << " classes" << std::endl;
    return 1;
  }
  std::cout << "PASS : ensemble image" << std::endl;
  return 0;
}
