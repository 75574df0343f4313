// KServe-v2 HTTP/REST client over raw POSIX sockets.
//
// API-compatible with the reference's InferenceServerHttpClient
// (src/c++/library/http_client.h:105-) but the transport is written
// here directly: a keep-alive blocking-socket pool for sync calls and
// an epoll worker thread multiplexing non-blocking transfers for
// AsyncInfer — the from-scratch equivalent of the reference's
// curl-multi + poll/wakeup loop (http_client.cc:2249-2348).
//
// Not thread-safe for concurrent Infer() on one client object (same
// rule as the reference, http_client.h:90-94); AsyncInfer may be called
// from any thread.
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <thread>
#include <vector>

#include "client_amd/common.h"
#include "client_amd/json.h"

namespace client_amd {

using Headers = std::map<std::string, std::string>;
using Parameters = std::map<std::string, std::string>;

// HTTPS options (reference HttpSslOptions http_client.h:45-86; TLS via
// system OpenSSL — sync path only this round, AsyncInfer stays h2c).
struct HttpSslOptions {
  bool verify_peer = true;
  bool verify_host = true;
  std::string ca_info;     // CA bundle path ("" = system default)
  std::string cert;        // client cert path (PEM)
  std::string key;         // client key path (PEM)
};

class InferResultHttp;

class InferenceServerHttpClient : public InferenceServerClient {
 public:
  enum class CompressionType { NONE, DEFLATE, GZIP };

  ~InferenceServerHttpClient() override;

  // url is host:port (no scheme), like the reference.
  static Error Create(
      std::unique_ptr<InferenceServerHttpClient>* client,
      const std::string& server_url, bool verbose = false);
  // HTTPS variant (reference Create overload with HttpSslOptions)
  static Error Create(
      std::unique_ptr<InferenceServerHttpClient>* client,
      const std::string& server_url, bool verbose, bool use_ssl,
      const HttpSslOptions& ssl_options);

  // ---- health / metadata ----
  Error IsServerLive(bool* live, const Headers& headers = {});
  Error IsServerReady(bool* ready, const Headers& headers = {});
  Error IsModelReady(
      bool* ready, const std::string& model_name,
      const std::string& model_version = "", const Headers& headers = {});
  Error ServerMetadata(std::string* server_metadata, const Headers& headers = {});
  Error ModelMetadata(
      std::string* model_metadata, const std::string& model_name,
      const std::string& model_version = "", const Headers& headers = {});
  Error ModelConfig(
      std::string* model_config, const std::string& model_name,
      const std::string& model_version = "", const Headers& headers = {});

  // ---- repository ----
  Error ModelRepositoryIndex(
      std::string* repository_index, const Headers& headers = {});
  Error LoadModel(
      const std::string& model_name, const Headers& headers = {},
      const std::string& config = "",
      const std::map<std::string, std::vector<char>>& files = {});
  Error UnloadModel(
      const std::string& model_name, const Headers& headers = {});

  // ---- statistics / trace / logging ----
  Error ModelInferenceStatistics(
      std::string* infer_stat, const std::string& model_name = "",
      const std::string& model_version = "", const Headers& headers = {});
  Error UpdateTraceSettings(
      std::string* response, const std::string& model_name = "",
      const std::map<std::string, std::vector<std::string>>& settings = {},
      const Headers& headers = {});
  Error GetTraceSettings(
      std::string* settings, const std::string& model_name = "",
      const Headers& headers = {});
  Error UpdateLogSettings(
      std::string* response, const std::map<std::string, std::string>& settings,
      const Headers& headers = {});
  Error GetLogSettings(std::string* settings, const Headers& headers = {});

  // ---- shared memory ----
  Error SystemSharedMemoryStatus(
      std::string* status, const std::string& region_name = "",
      const Headers& headers = {});
  Error RegisterSystemSharedMemory(
      const std::string& name, const std::string& key, size_t byte_size,
      size_t offset = 0, const Headers& headers = {});
  Error UnregisterSystemSharedMemory(
      const std::string& name = "", const Headers& headers = {});
  Error CudaSharedMemoryStatus(
      std::string* status, const std::string& region_name = "",
      const Headers& headers = {});
  // raw_handle: the 64 raw bytes of the hipIpcMemHandle_t (base64 on
  // the wire, reference http_client.cc:1708-1748)
  Error RegisterCudaSharedMemory(
      const std::string& name, const std::string& raw_handle,
      size_t device_id, size_t byte_size, const Headers& headers = {});
  Error UnregisterCudaSharedMemory(
      const std::string& name = "", const Headers& headers = {});
  // AMD-native spelling (same wire endpoint)
  Error RegisterHipSharedMemory(
      const std::string& name, const std::string& raw_handle,
      size_t device_id, size_t byte_size, const Headers& headers = {}) {
    return RegisterCudaSharedMemory(name, raw_handle, device_id, byte_size,
                                    headers);
  }

  // ---- inference ----
  Error Infer(
      InferResult** result, const InferOptions& options,
      const std::vector<InferInput*>& inputs,
      const std::vector<const InferRequestedOutput*>& outputs = {},
      const Headers& headers = {}, const Parameters& query_params = {},
      const CompressionType request_compression_algorithm =
          CompressionType::NONE,
      const CompressionType response_compression_algorithm =
          CompressionType::NONE);

  Error AsyncInfer(
      OnCompleteFn callback, const InferOptions& options,
      const std::vector<InferInput*>& inputs,
      const std::vector<const InferRequestedOutput*>& outputs = {},
      const Headers& headers = {}, const Parameters& query_params = {});

  Error InferMulti(
      std::vector<InferResult*>* results,
      const std::vector<InferOptions>& options,
      const std::vector<std::vector<InferInput*>>& inputs,
      const std::vector<std::vector<const InferRequestedOutput*>>& outputs =
          {},
      const Headers& headers = {}, const Parameters& query_params = {});

  Error AsyncInferMulti(
      OnMultiCompleteFn callback, const std::vector<InferOptions>& options,
      const std::vector<std::vector<InferInput*>>& inputs,
      const std::vector<std::vector<const InferRequestedOutput*>>& outputs =
          {},
      const Headers& headers = {}, const Parameters& query_params = {});

  // Stateless helpers for out-of-band use (perf tooling; reference
  // http_client.cc:1286-1351).
  static Error GenerateRequestBody(
      std::vector<char>* request_body, size_t* header_length,
      const InferOptions& options, const std::vector<InferInput*>& inputs,
      const std::vector<const InferRequestedOutput*>& outputs = {});
  static Error ParseResponseBody(
      InferResult** result, const std::vector<char>& response_body,
      size_t header_length);

 private:
  InferenceServerHttpClient(const std::string& url, bool verbose);

  Error DoRequest(
      int* http_code, std::string* response_body, const std::string& method,
      const std::string& path, const std::string& body,
      const Headers& headers, long timeout_us = 0,
      Headers* response_headers = nullptr);
  Error Get(
      std::string* response, const std::string& path,
      const Headers& headers, bool* ok_flag = nullptr);
  Error Post(
      std::string* response, const std::string& path,
      const std::string& body, const Headers& headers);

  void AsyncWorker();

  std::string host_;
  int port_;

  // sync connection (reused, recreated on failure)
  int sync_fd_ = -1;
  // TLS state for the sync path (OpenSSL; null when plain HTTP)
  bool use_ssl_ = false;
  HttpSslOptions ssl_options_;
  void* ssl_ctx_ = nullptr;  // SSL_CTX*
  void* ssl_ = nullptr;      // SSL* bound to sync_fd_
  Error EnsureSslCtx();
  Error NewSsl(int fd, void** ssl_out);  // SSL* with SNI + host checks
  Error SslConnect();
  void SslClose();
  bool IoSend(const char* data, size_t n);
  long IoRecv(char* buf, size_t n);

  // async machinery
  struct AsyncTransfer;
  std::thread worker_;
  std::mutex mu_;
  std::deque<std::unique_ptr<AsyncTransfer>> new_transfers_;
  int wakeup_fds_[2] = {-1, -1};
  std::atomic<bool> worker_running_{false};
};

//==============================================================================
// HTTP inference result (reference InferResultHttp http_client.cc:1042-).
//
class InferResultHttp : public InferResult {
 public:
  static void Create(
      InferResult** result, std::shared_ptr<std::string> response_body,
      size_t json_size, int http_code);

  Error ModelName(std::string* name) const override;
  Error ModelVersion(std::string* version) const override;
  Error Id(std::string* id) const override;
  Error Shape(
      const std::string& output_name,
      std::vector<int64_t>* shape) const override;
  Error Datatype(
      const std::string& output_name, std::string* datatype) const override;
  Error RawData(
      const std::string& output_name, const uint8_t** buf,
      size_t* byte_size) const override;
  std::string DebugString() const override { return response_json_.Dump(); }
  Error RequestStatus() const override { return status_; }

 private:
  InferResultHttp(
      std::shared_ptr<std::string> response_body, size_t json_size,
      int http_code);
  const Json* FindOutput(const std::string& name) const;

  std::shared_ptr<std::string> response_body_;
  Json response_json_;
  Error status_;
  // output name -> (offset into binary tail, size)
  std::map<std::string, std::pair<size_t, size_t>> binary_offsets_;
  size_t binary_base_ = 0;
};

}  // namespace client_amd
