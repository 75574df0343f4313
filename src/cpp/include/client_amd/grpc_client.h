// KServe-v2 gRPC client over the from-scratch h2 transport.
//
// API-compatible with the reference's InferenceServerGrpcClient
// (src/c++/library/grpc_client.h:100-) but with no grpc++/protobuf
// dependency: messages are hand-encoded (kserve_pb.h) and the wire is
// the local HTTP/2 implementation (h2.h) + gRPC framing (1-byte flag +
// 4-byte BE length per message). Supports sync Infer, callback
// AsyncInfer, and the bi-di ModelStreamInfer stream with decoupled
// final-response semantics (reference grpc_client.cc:1323-1416).
#pragma once

#include <memory>
#include <mutex>

#include "client_amd/common.h"
#include "client_amd/h2.h"
#include "client_amd/kserve_pb.h"

namespace client_amd {

class InferResultGrpc;

// TLS options, reference grpc_client.h SslOptions:112-136 — fields are
// PEM *contents* (read the files yourself, as the reference examples do).
struct SslOptions {
  std::string root_certificates;  // CA bundle PEM; empty = system roots
  std::string private_key;        // client key PEM (mTLS)
  std::string certificate_chain;  // client cert PEM (mTLS)
};

// gRPC keepalive knobs, reference grpc_client.h KeepAliveOptions:62-83
// (grpc/doc/keepalive.md). Default keepalive_time_ms of INT32_MAX
// means "never ping" — same as grpc-core.
struct KeepAliveOptions {
  int keepalive_time_ms = 0x7FFFFFFF;
  int keepalive_timeout_ms = 20000;
  bool keepalive_permit_without_calls = false;
  int http2_max_pings_without_data = 2;
};

class InferenceServerGrpcClient : public InferenceServerClient {
 public:
  ~InferenceServerGrpcClient() override;

  // Reference grpc_client.h:120-126 signature. TLS = OpenSSL + ALPN h2
  // with hostname/IP-SAN verification; keepalive = h2 PING watchdog;
  // use_cached_channel=false forces a private (unshared) connection.
  static Error Create(
      std::unique_ptr<InferenceServerGrpcClient>* client,
      const std::string& server_url, bool verbose = false,
      bool use_ssl = false, const SslOptions& ssl_options = SslOptions(),
      const KeepAliveOptions& keepalive_options = KeepAliveOptions(),
      const bool use_cached_channel = true);

  Error IsServerLive(bool* live);
  Error IsServerReady(bool* ready);
  Error IsModelReady(
      bool* ready, const std::string& model_name,
      const std::string& model_version = "");
  Error ServerMetadata(kserve::ServerMetadataPb* metadata);
  Error ModelMetadata(
      kserve::ModelMetadataPb* metadata, const std::string& model_name,
      const std::string& model_version = "");
  Error ModelConfig(
      kserve::ModelConfigPb* config, const std::string& model_name,
      const std::string& model_version = "");
  Error ModelRepositoryIndex(
      std::vector<kserve::RepositoryIndexEntryPb>* index);
  // optional config-override JSON + file-override blobs (reference
  // grpc_client.h LoadModel signature)
  Error LoadModel(
      const std::string& model_name, const std::string& config = "",
      const std::map<std::string, std::vector<char>>& files = {});
  Error UnloadModel(const std::string& model_name);
  // Reference grpc_client.h:330-352 trace-settings RPCs (the reference
  // has no gRPC log-settings; ours adds none either — Python covers it).
  Error UpdateTraceSettings(
      kserve::TraceSettingsPb* response, const std::string& model_name = "",
      const kserve::TraceSettingsPb& settings = {});
  Error GetTraceSettings(
      kserve::TraceSettingsPb* settings, const std::string& model_name = "");

  Error ModelInferenceStatistics(
      std::vector<kserve::ModelStatisticsPb>* stats,
      const std::string& model_name = "", const std::string& version = "");

  Error RegisterSystemSharedMemory(
      const std::string& name, const std::string& key, size_t byte_size,
      size_t offset = 0);
  Error UnregisterSystemSharedMemory(const std::string& name = "");
  // raw 64-byte hipIpcMemHandle_t bytes on the wire
  // (grpc_service.proto:1610-1643)
  Error RegisterCudaSharedMemory(
      const std::string& name, const std::string& raw_handle,
      size_t device_id, size_t byte_size);
  Error UnregisterCudaSharedMemory(const std::string& name = "");
  Error RegisterHipSharedMemory(
      const std::string& name, const std::string& raw_handle,
      size_t device_id, size_t byte_size) {
    return RegisterCudaSharedMemory(name, raw_handle, device_id, byte_size);
  }

  Error Infer(
      InferResult** result, const InferOptions& options,
      const std::vector<InferInput*>& inputs,
      const std::vector<const InferRequestedOutput*>& outputs = {});

  Error AsyncInfer(
      OnCompleteFn callback, const InferOptions& options,
      const std::vector<InferInput*>& inputs,
      const std::vector<const InferRequestedOutput*>& outputs = {});

  Error InferMulti(
      std::vector<InferResult*>* results,
      const std::vector<InferOptions>& options,
      const std::vector<std::vector<InferInput*>>& inputs,
      const std::vector<std::vector<const InferRequestedOutput*>>& outputs =
          {});
  Error AsyncInferMulti(
      OnMultiCompleteFn callback, const std::vector<InferOptions>& options,
      const std::vector<std::vector<InferInput*>>& inputs,
      const std::vector<std::vector<const InferRequestedOutput*>>& outputs =
          {});

  // Bi-di streaming: one active stream per client (reference rule).
  Error StartStream(OnCompleteFn stream_callback);
  Error AsyncStreamInfer(
      const InferOptions& options, const std::vector<InferInput*>& inputs,
      const std::vector<const InferRequestedOutput*>& outputs = {});
  Error StopStream();

 private:
  InferenceServerGrpcClient(const std::string& url, bool verbose);

  // Unary gRPC call: returns grpc-status + response bytes.
  Error UnaryCall(
      const std::string& method, const std::string& request,
      std::string* response, uint64_t timeout_us = 0);
  Error AsyncUnaryCall(
      const std::string& method, const std::string& request,
      std::function<void(Error, std::string)> on_done,
      uint64_t timeout_us = 0);
  kserve::ModelInferRequestPb BuildRequest(
      const InferOptions& options, const std::vector<InferInput*>& inputs,
      const std::vector<const InferRequestedOutput*>& outputs);
  Error EnsureConnected();

  std::string host_;
  int port_;
  H2SslOptions ssl_;
  H2KeepAlive keepalive_;
  bool use_cached_channel_ = true;
  // Shared h2 connection from the global per-url cache (reference
  // grpc_client.cc:80-152: up to TRITON_CLIENT_GRPC_CHANNEL_MAX_SHARE_COUNT
  // clients share one channel before a new one is created; HTTP/2
  // stream multiplexing makes sharing natural).
  std::shared_ptr<H2Connection> conn_;
  std::mutex conn_mu_;

  // active bidi stream state
  struct BidiState;
  std::shared_ptr<BidiState> bidi_;
};

//==============================================================================
// gRPC inference result: zero-copy views into raw_output_contents
// (reference InferResultGrpc grpc_client.cc:399-446).
class InferResultGrpc : public InferResult {
 public:
  static void Create(
      InferResult** result, std::shared_ptr<kserve::ModelInferResponsePb>
      response, Error status = Error::Success);

  Error ModelName(std::string* name) const override;
  Error ModelVersion(std::string* version) const override;
  Error Id(std::string* id) const override;
  Error Shape(const std::string& output_name,
              std::vector<int64_t>* shape) const override;
  Error Datatype(const std::string& output_name,
                 std::string* datatype) const override;
  Error RawData(const std::string& output_name, const uint8_t** buf,
                size_t* byte_size) const override;
  std::string DebugString() const override;
  Error RequestStatus() const override { return status_; }

  // decoupled-stream params (reference common.h:534-540)
  bool IsFinalResponse() const;
  bool IsNullResponse() const;

 private:
  InferResultGrpc(std::shared_ptr<kserve::ModelInferResponsePb> response,
                  Error status);
  const kserve::InferOutputTensorPb* Find(const std::string& name,
                                          size_t* raw_index) const;
  std::shared_ptr<kserve::ModelInferResponsePb> response_;
  Error status_;
};

}  // namespace client_amd
