// Image classification client (reference: src/c++/examples/image_client.cc,
// sans OpenCV): loads a binary PPM (P6) or synthesizes an image, does the
// same bilinear resize + scaling the server's HIP preprocess kernel does,
// converts to the model's wire dtype (FP32 or BF16 truncation), and uses
// the classification extension (class_count) to print top-k
// "<score>:<index>" results.
// Usage: image_client [-u host:port] [-m model] [-s NONE|INCEPTION|VGG]
//                     [-c topk] [--size N] [image.ppm]
#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
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

// Minimal P6 (binary RGB) PPM reader.
static bool LoadPpm(const std::string& path, std::vector<uint8_t>* pix,
                    int* h, int* w) {
  std::ifstream f(path, std::ios::binary);
  if (!f) return false;
  std::string magic;
  int maxval = 0;
  f >> magic >> *w >> *h >> maxval;
  if (magic != "P6" || maxval != 255) return false;
  f.get();  // single whitespace after header
  pix->resize((size_t)*h * *w * 3);
  f.read((char*)pix->data(), pix->size());
  return (bool)f;
}

// Bilinear resize HWC uint8 -> CHW fp32 + scaling (same math as the
// server's preprocess kernel; reference image_client preprocess).
static std::vector<float> Preprocess(const std::vector<uint8_t>& img, int ih,
                                     int iw, int size,
                                     const std::string& scaling) {
  std::vector<float> out((size_t)3 * size * size);
  const float sy = (float)ih / size, sx = (float)iw / size;
  for (int y = 0; y < size; ++y) {
    float fy = (y + 0.5f) * sy - 0.5f;
    int y0 = std::min(ih - 1, std::max(0, (int)std::floor(fy)));
    int y1 = std::min(ih - 1, y0 + 1);
    float wy = fy < 0 ? 0.f : fy - std::floor(fy);
    for (int x = 0; x < size; ++x) {
      float fx = (x + 0.5f) * sx - 0.5f;
      int x0 = std::min(iw - 1, std::max(0, (int)std::floor(fx)));
      int x1 = std::min(iw - 1, x0 + 1);
      float wx = fx < 0 ? 0.f : fx - std::floor(fx);
      for (int c = 0; c < 3; ++c) {
        auto px = [&](int yy, int xx) {
          return (float)img[((size_t)yy * iw + xx) * 3 + c];
        };
        float v = (1 - wy) * ((1 - wx) * px(y0, x0) + wx * px(y0, x1)) +
                  wy * ((1 - wx) * px(y1, x0) + wx * px(y1, x1));
        if (scaling == "INCEPTION") {
          v = v / 127.5f - 1.0f;
        } else if (scaling == "VGG") {
          static const float kMeans[3] = {104.f, 117.f, 123.f};
          v = v - kMeans[c];
        }
        out[((size_t)c * size + y) * size + x] = v;
      }
    }
  }
  return out;
}

int main(int argc, char** argv) {
  std::string url = "127.0.0.1:8000", model = "resnet50";
  std::string scaling = "INCEPTION", image_path;
  int topk = 3, size = 224;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "-u" && i + 1 < argc) url = argv[++i];
    else if (a == "-m" && i + 1 < argc) model = argv[++i];
    else if (a == "-s" && i + 1 < argc) scaling = argv[++i];
    else if (a == "-c" && i + 1 < argc) topk = atoi(argv[++i]);
    else if (a == "--size" && i + 1 < argc) size = atoi(argv[++i]);
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
    std::mt19937 rng(7);
    for (auto& b : img) b = (uint8_t)(rng() & 0xFF);
  }
  std::vector<float> chw = Preprocess(img, ih, iw, size, scaling);

  std::unique_ptr<ca::InferenceServerHttpClient> client;
  FAIL_IF_ERR(ca::InferenceServerHttpClient::Create(&client, url), "create");

  // wire dtype from model metadata (bf16 serving truncates fp32>>16)
  std::string meta;
  FAIL_IF_ERR(client->ModelMetadata(&meta, model), "metadata");
  bool bf16 = meta.find("\"BF16\"") != std::string::npos;

  ca::InferInput* input;
  FAIL_IF_ERR(ca::InferInput::Create(&input, "INPUT0", {1, 3, size, size},
                                     bf16 ? "BF16" : "FP32"),
              "INPUT0");
  std::unique_ptr<ca::InferInput> ip(input);
  std::vector<uint16_t> bf;
  if (bf16) {
    bf.resize(chw.size());
    for (size_t i = 0; i < chw.size(); ++i) {
      uint32_t u;
      memcpy(&u, &chw[i], 4);
      bf[i] = (uint16_t)(u >> 16);
    }
    FAIL_IF_ERR(input->AppendRaw((uint8_t*)bf.data(), bf.size() * 2), "set");
  } else {
    FAIL_IF_ERR(input->AppendRaw((uint8_t*)chw.data(), chw.size() * 4),
                "set");
  }

  ca::InferRequestedOutput* output;
  FAIL_IF_ERR(ca::InferRequestedOutput::Create(&output, "OUTPUT0", topk),
              "OUTPUT0");
  std::unique_ptr<ca::InferRequestedOutput> op(output);

  ca::InferOptions options(model);
  ca::InferResult* result;
  FAIL_IF_ERR(client->Infer(&result, options, {input}, {output}), "infer");
  std::unique_ptr<ca::InferResult> rp(result);
  std::vector<std::string> classes;
  FAIL_IF_ERR(result->StringData("OUTPUT0", &classes), "classes");
  for (const auto& c : classes) std::cout << "    " << c << std::endl;
  if ((int)classes.size() != topk) {
    std::cerr << "expected " << topk << " classes, got " << classes.size()
              << std::endl;
    return 1;
  }
  std::cout << "PASS : image classification" << std::endl;
  return 0;
}
