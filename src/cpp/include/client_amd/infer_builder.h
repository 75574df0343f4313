// Fluent inference-request builder — the C++ rendition of the
// reference Rust client's InferRequestBuilder ergonomic surface
// (reference src/rust/triton-client/src/infer.rs:548; typed
// with_data_* setters :210-447). Builds the (options, inputs, outputs)
// triple both clients' Infer/AsyncInfer take; owns the InferInput /
// InferRequestedOutput objects and the copied tensor bytes, so the
// caller only keeps the builder alive for the call.
//
//   InferRequestBuilder b("simple");
//   b.RequestId("r1")
//    .AddInput<int32_t>("INPUT0", {1, 16}, data0)
//    .AddInput<int32_t>("INPUT1", {1, 16}, data1)
//    .AddOutput("OUTPUT0");
//   client->Infer(&result, b.Options(), b.Inputs(), b.Outputs());

#pragma once

#include <cstring>
#include <memory>
#include <string>
#include <vector>

#include "client_amd/common.h"

namespace client_amd {

class InferRequestBuilder {
 public:
  explicit InferRequestBuilder(const std::string& model_name)
      : options_(model_name) {}

  InferRequestBuilder& ModelVersion(const std::string& v) {
    options_.model_version_ = v;
    return *this;
  }
  InferRequestBuilder& RequestId(const std::string& id) {
    options_.request_id_ = id;
    return *this;
  }
  InferRequestBuilder& Sequence(uint64_t id, bool start, bool end) {
    options_.sequence_id_ = id;
    options_.sequence_start_ = start;
    options_.sequence_end_ = end;
    return *this;
  }
  InferRequestBuilder& Priority(uint64_t p) {
    options_.priority_ = p;
    return *this;
  }
  InferRequestBuilder& ClientTimeout(uint64_t us) {
    options_.client_timeout_ = us;
    return *this;
  }

  // Typed tensor input: the data is COPIED into the builder (the Rust
  // builder owns its buffers the same way), so callers can pass
  // temporaries.
  template <typename T>
  InferRequestBuilder& AddInput(const std::string& name,
                                const std::vector<int64_t>& shape,
                                const std::vector<T>& data,
                                const std::string& datatype = "") {
    std::string dt = datatype.empty() ? DatatypeOf<T>() : datatype;
    InferInput* input = nullptr;
    InferInput::Create(&input, name, shape, dt);
    buffers_.emplace_back(
        reinterpret_cast<const char*>(data.data()),
        reinterpret_cast<const char*>(data.data()) + data.size() * sizeof(T));
    input->AppendRaw(
        reinterpret_cast<const uint8_t*>(buffers_.back().data()),
        buffers_.back().size());
    inputs_owned_.emplace_back(input);
    inputs_.push_back(input);
    return *this;
  }

  // BYTES input from strings (Rust with_data_bytes).
  InferRequestBuilder& AddBytesInput(const std::string& name,
                                     const std::vector<int64_t>& shape,
                                     const std::vector<std::string>& values) {
    InferInput* input = nullptr;
    InferInput::Create(&input, name, shape, "BYTES");
    input->AppendFromString(values);
    inputs_owned_.emplace_back(input);
    inputs_.push_back(input);
    return *this;
  }

  // Shared-memory input (no bytes on the wire).
  InferRequestBuilder& AddShmInput(const std::string& name,
                                   const std::vector<int64_t>& shape,
                                   const std::string& datatype,
                                   const std::string& region,
                                   size_t byte_size, size_t offset = 0) {
    InferInput* input = nullptr;
    InferInput::Create(&input, name, shape, datatype);
    input->SetSharedMemory(region, byte_size, offset);
    inputs_owned_.emplace_back(input);
    inputs_.push_back(input);
    return *this;
  }

  InferRequestBuilder& AddOutput(const std::string& name,
                                 size_t class_count = 0) {
    InferRequestedOutput* output = nullptr;
    InferRequestedOutput::Create(&output, name, class_count);
    outputs_owned_.emplace_back(output);
    outputs_.push_back(output);
    return *this;
  }

  InferRequestBuilder& AddShmOutput(const std::string& name,
                                    const std::string& region,
                                    size_t byte_size, size_t offset = 0) {
    InferRequestedOutput* output = nullptr;
    InferRequestedOutput::Create(&output, name);
    output->SetSharedMemory(region, byte_size, offset);
    outputs_owned_.emplace_back(output);
    outputs_.push_back(output);
    return *this;
  }

  const InferOptions& Options() const { return options_; }
  const std::vector<InferInput*>& Inputs() const { return inputs_; }
  const std::vector<const InferRequestedOutput*>& Outputs() const {
    return outputs_;
  }

  template <typename T>
  static std::string DatatypeOf();

 private:
  InferOptions options_;
  std::vector<std::string> buffers_;
  std::vector<std::unique_ptr<InferInput>> inputs_owned_;
  std::vector<std::unique_ptr<InferRequestedOutput>> outputs_owned_;
  std::vector<InferInput*> inputs_;
  std::vector<const InferRequestedOutput*> outputs_;
};

template <> inline std::string InferRequestBuilder::DatatypeOf<int8_t>() { return "INT8"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<int16_t>() { return "INT16"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<int32_t>() { return "INT32"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<int64_t>() { return "INT64"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<uint8_t>() { return "UINT8"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<uint16_t>() { return "UINT16"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<uint32_t>() { return "UINT32"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<uint64_t>() { return "UINT64"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<float>() { return "FP32"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<double>() { return "FP64"; }
template <> inline std::string InferRequestBuilder::DatatypeOf<bool>() { return "BOOL"; }

}  // namespace client_amd
