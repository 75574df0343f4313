#include "client_amd/grpc_client.h"

#include <condition_variable>
#include <cstring>
#include <deque>
#include <map>

namespace client_amd {

namespace {

constexpr const char* kService = "/inference.GRPCInferenceService/";

// Global per-url connection cache (reference grpc_client.cc:80-152).
struct SharedConn {
  std::shared_ptr<H2Connection> conn;
  int share_count = 0;
};
std::mutex g_conn_mu;
std::map<std::string, std::vector<SharedConn>> g_conns;

int MaxShareCount() {
  const char* env = getenv("TRITON_CLIENT_GRPC_CHANNEL_MAX_SHARE_COUNT");
  if (env != nullptr) {
    int v = atoi(env);
    if (v > 0) return v;
  }
  return 6;
}

std::shared_ptr<H2Connection> AcquireConnection(
    const std::string& host, int port, const H2SslOptions& ssl,
    const H2KeepAlive& ka, bool use_cache, Error* err) {
  // TLS and plaintext channels to the same endpoint must never share.
  std::string key = host + ":" + std::to_string(port) +
                    (ssl.use_ssl ? "+tls" : "");
  std::unique_lock<std::mutex> lock(g_conn_mu);
  if (use_cache) {
    auto& vec = g_conns[key];
    int max_share = MaxShareCount();
    for (auto& entry : vec) {
      if (entry.conn->IsAlive() && entry.share_count < max_share) {
        entry.share_count++;
        return entry.conn;
      }
    }
  }
  lock.unlock();
  auto conn = std::make_shared<H2Connection>();
  *err = conn->Connect(host, port, ssl);
  if (!err->IsOk()) return nullptr;
  conn->StartKeepalive(ka);
  if (use_cache) {
    lock.lock();
    g_conns[key].push_back({conn, 1});
  }
  return conn;
}

void ReleaseConnection(const std::string& host, int port, bool use_ssl,
                       const std::shared_ptr<H2Connection>& conn) {
  std::string key = host + ":" + std::to_string(port) +
                    (use_ssl ? "+tls" : "");
  std::lock_guard<std::mutex> lock(g_conn_mu);
  auto it = g_conns.find(key);
  if (it == g_conns.end()) {
    conn->Close();  // private (uncached) channel
    return;
  }
  auto& vec = it->second;
  for (size_t i = 0; i < vec.size(); ++i) {
    if (vec[i].conn == conn) {
      if (--vec[i].share_count <= 0) {
        vec[i].conn->Close();
        vec.erase(vec.begin() + i);
      }
      break;
    }
  }
}

// Accumulates DATA bytes and splits gRPC length-prefixed messages.
struct GrpcMessageBuffer {
  std::string buf;

  void Append(const uint8_t* data, size_t n) {
    buf.append((const char*)data, n);
  }

  // Returns true and fills msg when one complete message is buffered.
  bool Next(std::string* msg) {
    if (buf.size() < 5) return false;
    uint32_t len = ((uint32_t)(uint8_t)buf[1] << 24) |
                   ((uint32_t)(uint8_t)buf[2] << 16) |
                   ((uint32_t)(uint8_t)buf[3] << 8) | (uint8_t)buf[4];
    if (buf.size() < 5ull + len) return false;
    msg->assign(buf, 5, len);
    buf.erase(0, 5 + len);
    return true;
  }
};

std::string FrameGrpcMessage(const std::string& msg) {
  std::string out;
  out.reserve(5 + msg.size());
  out.push_back('\0');  // uncompressed
  uint32_t len = (uint32_t)msg.size();
  out.push_back((char)((len >> 24) & 0xFF));
  out.push_back((char)((len >> 16) & 0xFF));
  out.push_back((char)((len >> 8) & 0xFF));
  out.push_back((char)(len & 0xFF));
  out += msg;
  return out;
}

Error StatusFromTrailers(const HeaderList& trailers) {
  std::string status = "0";
  std::string message;
  for (const auto& kv : trailers) {
    if (kv.first == "grpc-status") status = kv.second;
    if (kv.first == "grpc-message") message = kv.second;
  }
  if (status == "0") return Error::Success;
  if (message.empty()) message = "grpc error status " + status;
  if (status == "4") message = "Deadline Exceeded: " + message;
  return Error(message);
}

std::string GrpcTimeoutValue(uint64_t timeout_us) {
  return std::to_string(timeout_us) + "u";
}

}  // namespace

//==============================================================================
// InferResultGrpc

void InferResultGrpc::Create(
    InferResult** result, std::shared_ptr<kserve::ModelInferResponsePb>
    response, Error status) {
  *result = new InferResultGrpc(std::move(response), std::move(status));
}

InferResultGrpc::InferResultGrpc(
    std::shared_ptr<kserve::ModelInferResponsePb> response, Error status)
    : response_(std::move(response)), status_(std::move(status)) {}

const kserve::InferOutputTensorPb* InferResultGrpc::Find(
    const std::string& name, size_t* raw_index) const {
  size_t idx = 0;
  for (const auto& out : response_->outputs) {
    bool shm = out.parameters.count("shared_memory_region") > 0;
    if (out.name == name) {
      *raw_index = shm ? SIZE_MAX : idx;
      return &out;
    }
    if (!shm) idx++;
  }
  return nullptr;
}

Error InferResultGrpc::ModelName(std::string* name) const {
  *name = response_->model_name;
  return Error::Success;
}

Error InferResultGrpc::ModelVersion(std::string* version) const {
  *version = response_->model_version;
  return Error::Success;
}

Error InferResultGrpc::Id(std::string* id) const {
  *id = response_->id;
  return Error::Success;
}

Error InferResultGrpc::Shape(
    const std::string& output_name, std::vector<int64_t>* shape) const {
  size_t idx;
  const auto* out = Find(output_name, &idx);
  if (out == nullptr)
    return Error("no result found for requested output: " + output_name);
  *shape = out->shape;
  return Error::Success;
}

Error InferResultGrpc::Datatype(
    const std::string& output_name, std::string* datatype) const {
  size_t idx;
  const auto* out = Find(output_name, &idx);
  if (out == nullptr)
    return Error("no result found for requested output: " + output_name);
  *datatype = out->datatype;
  return Error::Success;
}

Error InferResultGrpc::RawData(
    const std::string& output_name, const uint8_t** buf,
    size_t* byte_size) const {
  size_t idx;
  const auto* out = Find(output_name, &idx);
  if (out == nullptr)
    return Error("no result found for requested output: " + output_name);
  if (idx == SIZE_MAX || idx >= response_->raw_output_contents.size())
    return Error("no raw data for output: " + output_name);
  const std::string& raw = response_->raw_output_contents[idx];
  *buf = (const uint8_t*)raw.data();
  *byte_size = raw.size();
  return Error::Success;
}

std::string InferResultGrpc::DebugString() const {
  return "ModelInferResponse{model=" + response_->model_name +
         ", id=" + response_->id +
         ", outputs=" + std::to_string(response_->outputs.size()) + "}";
}

bool InferResultGrpc::IsFinalResponse() const {
  auto it = response_->parameters.find("triton_final_response");
  return it != response_->parameters.end() && it->second.b;
}

bool InferResultGrpc::IsNullResponse() const {
  auto it = response_->parameters.find("triton_null_response");
  return it != response_->parameters.end() && it->second.b;
}

//==============================================================================
// InferenceServerGrpcClient

struct InferenceServerGrpcClient::BidiState {
  int32_t stream_id = -1;
  GrpcMessageBuffer messages;
  OnCompleteFn callback;
  std::mutex mu;
  bool done = false;
};

Error InferenceServerGrpcClient::Create(
    std::unique_ptr<InferenceServerGrpcClient>* client,
    const std::string& server_url, bool verbose, bool use_ssl,
    const SslOptions& ssl_options, const KeepAliveOptions& keepalive_options,
    const bool use_cached_channel) {
  client->reset(new InferenceServerGrpcClient(server_url, verbose));
  (*client)->ssl_.use_ssl = use_ssl;
  (*client)->ssl_.root_certificates = ssl_options.root_certificates;
  (*client)->ssl_.private_key = ssl_options.private_key;
  (*client)->ssl_.certificate_chain = ssl_options.certificate_chain;
  // INT32_MAX = grpc-core's "keepalive off" sentinel
  (*client)->keepalive_.enabled =
      keepalive_options.keepalive_time_ms != 0x7FFFFFFF;
  (*client)->keepalive_.time_ms = keepalive_options.keepalive_time_ms;
  (*client)->keepalive_.timeout_ms = keepalive_options.keepalive_timeout_ms;
  (*client)->keepalive_.permit_without_calls =
      keepalive_options.keepalive_permit_without_calls;
  (*client)->keepalive_.max_pings_without_data =
      keepalive_options.http2_max_pings_without_data;
  (*client)->use_cached_channel_ = use_cached_channel;
  return Error::Success;
}

InferenceServerGrpcClient::InferenceServerGrpcClient(
    const std::string& url, bool verbose)
    : InferenceServerClient(verbose) {
  size_t colon = url.rfind(':');
  if (colon == std::string::npos) {
    host_ = url;
    port_ = 8001;
  } else {
    host_ = url.substr(0, colon);
    port_ = atoi(url.c_str() + colon + 1);
  }
}

InferenceServerGrpcClient::~InferenceServerGrpcClient() {
  StopStream();
  if (conn_ != nullptr) {
    ReleaseConnection(host_, port_, ssl_.use_ssl, conn_);
    conn_.reset();
  }
}

Error InferenceServerGrpcClient::EnsureConnected() {
  std::lock_guard<std::mutex> lock(conn_mu_);
  if (conn_ != nullptr && conn_->IsAlive()) return Error::Success;
  if (conn_ != nullptr) {
    ReleaseConnection(host_, port_, ssl_.use_ssl, conn_);
    conn_.reset();
  }
  Error err = Error::Success;
  conn_ = AcquireConnection(host_, port_, ssl_, keepalive_,
                            use_cached_channel_, &err);
  return err;
}

Error InferenceServerGrpcClient::AsyncUnaryCall(
    const std::string& method, const std::string& request,
    std::function<void(Error, std::string)> on_done, uint64_t timeout_us) {
  RETURN_IF_ERROR(EnsureConnected());

  HeaderList headers = {
      {":method", "POST"},
      {":scheme", ssl_.use_ssl ? "https" : "http"},
      {":path", std::string(kService) + method},
      {":authority", host_ + ":" + std::to_string(port_)},
      {"te", "trailers"},
      {"content-type", "application/grpc"},
      {"user-agent", "client-amd-cpp/0.1"},
  };
  if (timeout_us > 0) {
    headers.emplace_back("grpc-timeout", GrpcTimeoutValue(timeout_us));
  }

  struct CallState {
    GrpcMessageBuffer messages;
    std::string response;
    bool got_message = false;
    std::function<void(Error, std::string)> on_done;
    std::mutex mu;
    bool finished = false;
  };
  auto state = std::make_shared<CallState>();
  state->on_done = std::move(on_done);

  auto finish = [state](Error err) {
    std::function<void(Error, std::string)> cb;
    std::string resp;
    {
      std::lock_guard<std::mutex> lock(state->mu);
      if (state->finished) return;
      state->finished = true;
      cb = state->on_done;
      resp = std::move(state->response);
    }
    cb(std::move(err), std::move(resp));
  };

  H2Connection::StreamHandler handler;
  handler.on_data = [state](const uint8_t* data, size_t n) {
    std::lock_guard<std::mutex> lock(state->mu);
    state->messages.Append(data, n);
    std::string msg;
    while (state->messages.Next(&msg)) {
      state->response = std::move(msg);
      state->got_message = true;
    }
  };
  handler.on_headers = [state, finish](const HeaderList& headers) {
    for (const auto& kv : headers) {
      // trailers-only responses put grpc-status in initial headers
      if (kv.first == "grpc-status" && kv.second != "0") {
        finish(StatusFromTrailers(headers));
        return;
      }
    }
  };
  handler.on_trailers = [state, finish](const HeaderList& trailers) {
    Error status = StatusFromTrailers(trailers);
    bool got;
    {
      std::lock_guard<std::mutex> lock(state->mu);
      got = state->got_message;
    }
    if (status.IsOk() && !got) {
      status = Error("gRPC call completed without a response message");
    }
    finish(status);
  };
  handler.on_error = [finish](const std::string& msg) { finish(Error(msg)); };

  int32_t stream_id;
  {
    std::lock_guard<std::mutex> lock(conn_mu_);
    RETURN_IF_ERROR(conn_->StartStream(headers, handler, &stream_id));
    std::string framed = FrameGrpcMessage(request);
    RETURN_IF_ERROR(conn_->SendData(
        stream_id, (const uint8_t*)framed.data(), framed.size(), true));
  }
  return Error::Success;
}

Error InferenceServerGrpcClient::UnaryCall(
    const std::string& method, const std::string& request,
    std::string* response, uint64_t timeout_us) {
  // Heap-allocated wait state: the completion callback may retain it
  // past this frame on error paths, and fresh heap addresses keep
  // TSAN's mutex shadow clean (stack-reused std::mutex never runs
  // pthread_mutex_destroy and poisons later reports).
  struct WaitState {
    std::mutex mu;
    std::condition_variable cv;
    bool done = false;
    Error err{""};
    std::string body;
  };
  auto st = std::make_shared<WaitState>();
  RETURN_IF_ERROR(AsyncUnaryCall(
      method, request,
      [st](Error err, std::string body) {
        std::lock_guard<std::mutex> lock(st->mu);
        st->err = std::move(err);
        st->body = std::move(body);
        st->done = true;
        st->cv.notify_all();
      },
      timeout_us));
  std::unique_lock<std::mutex> lock(st->mu);
  st->cv.wait(lock, [&] { return st->done; });
  RETURN_IF_ERROR(st->err);
  *response = std::move(st->body);
  return Error::Success;
}

// ---- management RPCs ----

Error InferenceServerGrpcClient::IsServerLive(bool* live) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall("ServerLive", kserve::EncodeEmpty(), &resp));
  *live = kserve::DecodeBoolField1((const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::IsServerReady(bool* ready) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall("ServerReady", kserve::EncodeEmpty(), &resp));
  *ready = kserve::DecodeBoolField1((const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::IsModelReady(
    bool* ready, const std::string& model_name,
    const std::string& model_version) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall(
      "ModelReady", kserve::EncodeNameVersion(model_name, model_version),
      &resp));
  *ready = kserve::DecodeBoolField1((const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::ServerMetadata(
    kserve::ServerMetadataPb* metadata) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall("ServerMetadata", kserve::EncodeEmpty(), &resp));
  *metadata = kserve::ServerMetadataPb::Decode(
      (const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::ModelMetadata(
    kserve::ModelMetadataPb* metadata, const std::string& model_name,
    const std::string& model_version) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall(
      "ModelMetadata", kserve::EncodeNameVersion(model_name, model_version),
      &resp));
  *metadata = kserve::ModelMetadataPb::Decode(
      (const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::ModelConfig(
    kserve::ModelConfigPb* config, const std::string& model_name,
    const std::string& model_version) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall(
      "ModelConfig", kserve::EncodeNameVersion(model_name, model_version),
      &resp));
  *config = kserve::ModelConfigPb::Decode(
      (const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::ModelRepositoryIndex(
    std::vector<kserve::RepositoryIndexEntryPb>* index) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall("RepositoryIndex", kserve::EncodeEmpty(), &resp));
  *index = kserve::DecodeRepositoryIndex(
      (const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::LoadModel(
    const std::string& model_name, const std::string& config,
    const std::map<std::string, std::vector<char>>& files) {
  std::map<std::string, std::string> file_blobs;
  for (const auto& [path, content] : files) {
    file_blobs.emplace(path,
                       std::string(content.data(), content.size()));
  }
  std::string resp;
  return UnaryCall(
      "RepositoryModelLoad",
      kserve::EncodeRepositoryModelLoadRequest(model_name, config,
                                               file_blobs),
      &resp);
}

Error InferenceServerGrpcClient::UnloadModel(const std::string& model_name) {
  std::string resp;
  return UnaryCall(
      "RepositoryModelUnload",
      kserve::EncodeRepositoryModelRequest(model_name), &resp);
}

Error InferenceServerGrpcClient::UpdateTraceSettings(
    kserve::TraceSettingsPb* response, const std::string& model_name,
    const kserve::TraceSettingsPb& settings) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall(
      "TraceSetting",
      kserve::EncodeTraceSettingRequest(settings, model_name), &resp));
  if (response != nullptr) {
    *response = kserve::DecodeTraceSettingResponse(
        (const uint8_t*)resp.data(), resp.size());
  }
  return Error::Success;
}

Error InferenceServerGrpcClient::GetTraceSettings(
    kserve::TraceSettingsPb* settings, const std::string& model_name) {
  return UpdateTraceSettings(settings, model_name, {});
}

Error InferenceServerGrpcClient::ModelInferenceStatistics(
    std::vector<kserve::ModelStatisticsPb>* stats,
    const std::string& model_name, const std::string& version) {
  std::string resp;
  RETURN_IF_ERROR(UnaryCall(
      "ModelStatistics", kserve::EncodeNameVersion(model_name, version),
      &resp));
  *stats = kserve::DecodeModelStatistics(
      (const uint8_t*)resp.data(), resp.size());
  return Error::Success;
}

Error InferenceServerGrpcClient::RegisterSystemSharedMemory(
    const std::string& name, const std::string& key, size_t byte_size,
    size_t offset) {
  std::string resp;
  return UnaryCall(
      "SystemSharedMemoryRegister",
      kserve::EncodeSystemShmRegister(name, key, offset, byte_size), &resp);
}

Error InferenceServerGrpcClient::UnregisterSystemSharedMemory(
    const std::string& name) {
  std::string resp;
  return UnaryCall(
      "SystemSharedMemoryUnregister", kserve::EncodeName(name), &resp);
}

Error InferenceServerGrpcClient::RegisterCudaSharedMemory(
    const std::string& name, const std::string& raw_handle, size_t device_id,
    size_t byte_size) {
  std::string resp;
  return UnaryCall(
      "CudaSharedMemoryRegister",
      kserve::EncodeCudaShmRegister(name, raw_handle, (int64_t)device_id,
                                    byte_size),
      &resp);
}

Error InferenceServerGrpcClient::UnregisterCudaSharedMemory(
    const std::string& name) {
  std::string resp;
  return UnaryCall(
      "CudaSharedMemoryUnregister", kserve::EncodeName(name), &resp);
}

// ---- inference ----

kserve::ModelInferRequestPb InferenceServerGrpcClient::BuildRequest(
    const InferOptions& options, const std::vector<InferInput*>& inputs,
    const std::vector<const InferRequestedOutput*>& outputs) {
  kserve::ModelInferRequestPb req;
  req.model_name = options.model_name_;
  req.model_version = options.model_version_;
  req.id = options.request_id_;
  if (options.sequence_id_ != 0) {
    req.parameters["sequence_id"] =
        kserve::InferParameter::Int((int64_t)options.sequence_id_);
    req.parameters["sequence_start"] =
        kserve::InferParameter::Bool(options.sequence_start_);
    req.parameters["sequence_end"] =
        kserve::InferParameter::Bool(options.sequence_end_);
  } else if (!options.sequence_id_str_.empty()) {
    req.parameters["sequence_id"] =
        kserve::InferParameter::Str(options.sequence_id_str_);
    req.parameters["sequence_start"] =
        kserve::InferParameter::Bool(options.sequence_start_);
    req.parameters["sequence_end"] =
        kserve::InferParameter::Bool(options.sequence_end_);
  }
  if (options.priority_ != 0) {
    req.parameters["priority"] =
        kserve::InferParameter::Uint(options.priority_);
  }
  if (options.server_timeout_ != 0) {
    req.parameters["timeout"] =
        kserve::InferParameter::Int((int64_t)options.server_timeout_);
  }
  if (options.triton_enable_empty_final_response_) {
    req.parameters["triton_enable_empty_final_response"] =
        kserve::InferParameter::Bool(true);
  }
  for (const auto& kv : options.request_parameters_) {
    req.parameters[kv.first] = kserve::InferParameter::Str(kv.second);
  }
  for (InferInput* input : inputs) {
    kserve::InferInputTensorPb t;
    t.name = input->Name();
    t.datatype = input->Datatype();
    t.shape = input->Shape();
    if (input->IsSharedMemory()) {
      std::string region;
      size_t size, offset;
      input->SharedMemoryInfo(&region, &size, &offset);
      t.parameters["shared_memory_region"] =
          kserve::InferParameter::Str(region);
      t.parameters["shared_memory_byte_size"] =
          kserve::InferParameter::Int((int64_t)size);
      if (offset != 0) {
        t.parameters["shared_memory_offset"] =
            kserve::InferParameter::Int((int64_t)offset);
      }
    } else {
      // gather the scatter list into raw_input_contents (the wire form
      // the protocol prefers, grpc_service.proto:683-706)
      input->PrepareForRequest();
      std::string raw;
      raw.reserve(input->ByteSize());
      const uint8_t* buf;
      size_t n;
      bool end = false;
      while (!end) {
        input->GetNext(&buf, &n, &end);
        if (buf != nullptr && n > 0) raw.append((const char*)buf, n);
      }
      req.raw_input_contents.push_back(std::move(raw));
    }
    req.inputs.push_back(std::move(t));
  }
  for (const InferRequestedOutput* output : outputs) {
    kserve::InferRequestedOutputPb t;
    t.name = output->Name();
    if (output->ClassCount() != 0) {
      t.parameters["classification"] =
          kserve::InferParameter::Int((int64_t)output->ClassCount());
    }
    if (output->IsSharedMemory()) {
      std::string region;
      size_t size, offset;
      output->SharedMemoryInfo(&region, &size, &offset);
      t.parameters["shared_memory_region"] =
          kserve::InferParameter::Str(region);
      t.parameters["shared_memory_byte_size"] =
          kserve::InferParameter::Int((int64_t)size);
      if (offset != 0) {
        t.parameters["shared_memory_offset"] =
            kserve::InferParameter::Int((int64_t)offset);
      }
    }
    req.outputs.push_back(std::move(t));
  }
  return req;
}

Error InferenceServerGrpcClient::AsyncInfer(
    OnCompleteFn callback, const InferOptions& options,
    const std::vector<InferInput*>& inputs,
    const std::vector<const InferRequestedOutput*>& outputs) {
  if (callback == nullptr)
    return Error("callback must not be null for AsyncInfer");
  auto timer = std::make_shared<RequestTimers>();
  timer->CaptureTimestamp(RequestTimers::Kind::REQUEST_START);
  timer->CaptureTimestamp(RequestTimers::Kind::SEND_START);
  std::string encoded = BuildRequest(options, inputs, outputs).Encode();
  timer->CaptureTimestamp(RequestTimers::Kind::SEND_END);
  return AsyncUnaryCall(
      "ModelInfer", encoded,
      [this, callback, timer](Error err, std::string body) {
        timer->CaptureTimestamp(RequestTimers::Kind::RECV_START);
        auto response = std::make_shared<kserve::ModelInferResponsePb>();
        if (err.IsOk()) {
          *response = kserve::ModelInferResponsePb::Decode(
              (const uint8_t*)body.data(), body.size());
        }
        timer->CaptureTimestamp(RequestTimers::Kind::RECV_END);
        timer->CaptureTimestamp(RequestTimers::Kind::REQUEST_END);
        UpdateInferStat(*timer);
        InferResult* result = nullptr;
        InferResultGrpc::Create(&result, std::move(response), std::move(err));
        callback(result);
      },
      options.client_timeout_);
}

Error InferenceServerGrpcClient::Infer(
    InferResult** result, const InferOptions& options,
    const std::vector<InferInput*>& inputs,
    const std::vector<const InferRequestedOutput*>& outputs) {
  struct WaitState {  // heap state, same rationale as UnaryCall
    std::mutex mu;
    std::condition_variable cv;
    InferResult* res = nullptr;
    bool done = false;
  };
  auto st = std::make_shared<WaitState>();
  RETURN_IF_ERROR(AsyncInfer(
      [st](InferResult* r) {
        std::lock_guard<std::mutex> lock(st->mu);
        st->res = r;
        st->done = true;
        st->cv.notify_all();
      },
      options, inputs, outputs));
  std::unique_lock<std::mutex> lock(st->mu);
  st->cv.wait(lock, [&] { return st->done; });
  *result = st->res;
  return st->res->RequestStatus();
}

Error InferenceServerGrpcClient::InferMulti(
    std::vector<InferResult*>* results,
    const std::vector<InferOptions>& options,
    const std::vector<std::vector<InferInput*>>& inputs,
    const std::vector<std::vector<const InferRequestedOutput*>>& outputs) {
  if (options.size() != 1 && options.size() != inputs.size()) {
    return Error("'options' must be of size 1 or match 'inputs'");
  }
  if (!outputs.empty() && outputs.size() != 1 &&
      outputs.size() != inputs.size()) {
    return Error("'outputs' must be empty, size 1, or match 'inputs'");
  }
  results->clear();
  for (size_t i = 0; i < inputs.size(); ++i) {
    const InferOptions& opt = options.size() == 1 ? options[0] : options[i];
    std::vector<const InferRequestedOutput*> outs;
    if (!outputs.empty())
      outs = outputs.size() == 1 ? outputs[0] : outputs[i];
    InferResult* result = nullptr;
    Error err = Infer(&result, opt, inputs[i], outs);
    results->push_back(result);
    if (!err.IsOk()) return err;
  }
  return Error::Success;
}

Error InferenceServerGrpcClient::AsyncInferMulti(
    OnMultiCompleteFn callback, const std::vector<InferOptions>& options,
    const std::vector<std::vector<InferInput*>>& inputs,
    const std::vector<std::vector<const InferRequestedOutput*>>& outputs) {
  if (options.size() != 1 && options.size() != inputs.size()) {
    return Error("'options' must be of size 1 or match 'inputs'");
  }
  if (!outputs.empty() && outputs.size() != 1 &&
      outputs.size() != inputs.size()) {
    return Error("'outputs' must be empty, size 1, or match 'inputs'");
  }
  // atomic countdown join -> single callback
  // (reference grpc_client.cc:1283-1302)
  struct MultiState {
    std::mutex mu;
    std::vector<InferResult*> results;
    size_t remaining;
    OnMultiCompleteFn callback;
  };
  auto state = std::make_shared<MultiState>();
  state->results.resize(inputs.size(), nullptr);
  state->remaining = inputs.size();
  state->callback = std::move(callback);
  for (size_t i = 0; i < inputs.size(); ++i) {
    const InferOptions& opt = options.size() == 1 ? options[0] : options[i];
    std::vector<const InferRequestedOutput*> outs;
    if (!outputs.empty())
      outs = outputs.size() == 1 ? outputs[0] : outputs[i];
    Error err = AsyncInfer(
        [state, i](InferResult* result) {
          bool fire = false;
          {
            std::lock_guard<std::mutex> lock(state->mu);
            state->results[i] = result;
            fire = (--state->remaining == 0);
          }
          if (fire) state->callback(state->results);
        },
        opt, inputs[i], outs);
    if (!err.IsOk()) return err;
  }
  return Error::Success;
}

// ---- bi-di streaming ----

Error InferenceServerGrpcClient::StartStream(OnCompleteFn stream_callback) {
  if (bidi_ != nullptr) {
    return Error(
        "cannot start another stream with one already running. "
        "'InferenceServerGrpcClient' supports only a single active "
        "stream at a given time.");
  }
  RETURN_IF_ERROR(EnsureConnected());
  auto state = std::make_shared<BidiState>();
  state->callback = std::move(stream_callback);

  H2Connection::StreamHandler handler;
  handler.on_data = [state](const uint8_t* data, size_t n) {
    std::vector<std::string> complete;
    {
      std::lock_guard<std::mutex> lock(state->mu);
      state->messages.Append(data, n);
      std::string msg;
      while (state->messages.Next(&msg)) complete.push_back(std::move(msg));
    }
    for (const auto& msg : complete) {
      auto stream_resp = kserve::ModelStreamInferResponsePb::Decode(
          (const uint8_t*)msg.data(), msg.size());
      auto response = std::make_shared<kserve::ModelInferResponsePb>(
          std::move(stream_resp.infer_response));
      Error status = stream_resp.error_message.empty()
                         ? Error::Success
                         : Error(stream_resp.error_message);
      InferResult* result = nullptr;
      InferResultGrpc::Create(&result, std::move(response),
                              std::move(status));
      state->callback(result);
    }
  };
  handler.on_trailers = [state](const HeaderList& trailers) {
    Error status = StatusFromTrailers(trailers);
    std::lock_guard<std::mutex> lock(state->mu);
    state->done = true;
    if (!status.IsOk()) {
      InferResult* result = nullptr;
      InferResultGrpc::Create(
          &result, std::make_shared<kserve::ModelInferResponsePb>(), status);
      state->callback(result);
    }
  };
  handler.on_error = [state](const std::string& msg) {
    std::lock_guard<std::mutex> lock(state->mu);
    if (state->done) return;
    state->done = true;
    InferResult* result = nullptr;
    InferResultGrpc::Create(
        &result, std::make_shared<kserve::ModelInferResponsePb>(),
        Error(msg));
    state->callback(result);
  };

  HeaderList headers = {
      {":method", "POST"},
      {":scheme", ssl_.use_ssl ? "https" : "http"},
      {":path", std::string(kService) + "ModelStreamInfer"},
      {":authority", host_ + ":" + std::to_string(port_)},
      {"te", "trailers"},
      {"content-type", "application/grpc"},
      {"user-agent", "client-amd-cpp/0.1"},
  };
  int32_t stream_id;
  {
    std::lock_guard<std::mutex> lock(conn_mu_);
    RETURN_IF_ERROR(conn_->StartStream(headers, handler, &stream_id));
  }
  state->stream_id = stream_id;
  bidi_ = state;
  return Error::Success;
}

Error InferenceServerGrpcClient::AsyncStreamInfer(
    const InferOptions& options, const std::vector<InferInput*>& inputs,
    const std::vector<const InferRequestedOutput*>& outputs) {
  if (bidi_ == nullptr) {
    return Error("stream not available, use StartStream() to make one");
  }
  std::string framed = FrameGrpcMessage(
      BuildRequest(options, inputs, outputs).Encode());
  std::lock_guard<std::mutex> lock(conn_mu_);
  return conn_->SendData(
      bidi_->stream_id, (const uint8_t*)framed.data(), framed.size(), false);
}

Error InferenceServerGrpcClient::StopStream() {
  if (bidi_ == nullptr) return Error::Success;
  auto state = bidi_;
  bidi_ = nullptr;
  {
    std::lock_guard<std::mutex> lock(conn_mu_);
    if (conn_ != nullptr && conn_->IsAlive()) {
      conn_->FinishStream(state->stream_id);
    }
  }
  // brief wait for the server to close its half
  for (int i = 0; i < 100; ++i) {
    {
      std::lock_guard<std::mutex> lock(state->mu);
      if (state->done) break;
    }
    struct timespec ts {0, 10 * 1000 * 1000};
    nanosleep(&ts, nullptr);
  }
  return Error::Success;
}

}  // namespace client_amd
