#include "client_amd/http_client.h"

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <signal.h>
#include <string.h>
#include <sys/epoll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <mutex>
#include <sstream>

#include <openssl/err.h>
#include <openssl/ssl.h>
#include <zlib.h>

#include "client_amd/base64.h"

namespace client_amd {

namespace {

constexpr size_t kRecvChunk = 1 << 20;  // 1 MiB recv buffer

std::string UrlEncode(const std::string& s) {
  static const char* hex = "0123456789ABCDEF";
  std::string out;
  for (unsigned char c : s) {
    if (isalnum(c) || c == '-' || c == '_' || c == '.' || c == '~') {
      out.push_back((char)c);
    } else {
      out.push_back('%');
      out.push_back(hex[c >> 4]);
      out.push_back(hex[c & 15]);
    }
  }
  return out;
}

int ConnectTo(const std::string& host, int port, bool nonblocking) {
  struct addrinfo hints;
  memset(&hints, 0, sizeof(hints));
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  std::string port_str = std::to_string(port);
  if (getaddrinfo(host.c_str(), port_str.c_str(), &hints, &res) != 0) {
    return -1;
  }
  int fd = -1;
  for (struct addrinfo* rp = res; rp != nullptr; rp = rp->ai_next) {
    fd = socket(rp->ai_family, rp->ai_socktype, rp->ai_protocol);
    if (fd < 0) continue;
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    if (nonblocking) {
      fcntl(fd, F_SETFL, fcntl(fd, F_GETFL, 0) | O_NONBLOCK);
    }
    if (connect(fd, rp->ai_addr, rp->ai_addrlen) == 0 ||
        (nonblocking && errno == EINPROGRESS)) {
      break;
    }
    close(fd);
    fd = -1;
  }
  freeaddrinfo(res);
  return fd;
}

bool SendAll(int fd, const char* data, size_t len) {
  size_t sent = 0;
  while (sent < len) {
    ssize_t n = send(fd, data + sent, len - sent, MSG_NOSIGNAL);
    if (n <= 0) {
      if (n < 0 && (errno == EINTR)) continue;
      return false;
    }
    sent += (size_t)n;
  }
  return true;
}

// Parse response head in `buf` ending at header_end (index of the
// "\r\n\r\n"). Returns http status; fills headers (lower-cased keys).
int ParseResponseHead(
    const std::string& buf, size_t header_end, Headers* headers) {
  size_t line_end = buf.find("\r\n");
  if (line_end == std::string::npos) return -1;
  // "HTTP/1.1 200 OK"
  size_t sp1 = buf.find(' ');
  if (sp1 == std::string::npos || sp1 >= line_end) return -1;
  int code = atoi(buf.c_str() + sp1 + 1);
  size_t pos = line_end + 2;
  while (pos < header_end) {
    size_t eol = buf.find("\r\n", pos);
    if (eol == std::string::npos || eol > header_end) eol = header_end;
    size_t colon = buf.find(':', pos);
    if (colon != std::string::npos && colon < eol) {
      std::string key = buf.substr(pos, colon - pos);
      for (auto& c : key) c = (char)tolower(c);
      size_t vstart = colon + 1;
      while (vstart < eol && buf[vstart] == ' ') vstart++;
      (*headers)[key] = buf.substr(vstart, eol - vstart);
    }
    pos = eol + 2;
  }
  return code;
}

void AppendDataJson(
    JsonArray* data, const std::string& datatype, const uint8_t* buf,
    size_t nbytes) {
  // binary -> JSON "data" conversion for non-binary transport
  // (reference ConvertBinaryInputsToJSON http_client.cc:606-678)
  if (datatype == "FP32") {
    const float* v = (const float*)buf;
    for (size_t i = 0; i < nbytes / 4; ++i) data->push_back(Json((double)v[i]));
  } else if (datatype == "FP64") {
    const double* v = (const double*)buf;
    for (size_t i = 0; i < nbytes / 8; ++i) data->push_back(Json(v[i]));
  } else if (datatype == "INT32") {
    const int32_t* v = (const int32_t*)buf;
    for (size_t i = 0; i < nbytes / 4; ++i)
      data->push_back(Json((int64_t)v[i]));
  } else if (datatype == "INT64") {
    const int64_t* v = (const int64_t*)buf;
    for (size_t i = 0; i < nbytes / 8; ++i) data->push_back(Json(v[i]));
  } else if (datatype == "INT16") {
    const int16_t* v = (const int16_t*)buf;
    for (size_t i = 0; i < nbytes / 2; ++i)
      data->push_back(Json((int64_t)v[i]));
  } else if (datatype == "INT8") {
    const int8_t* v = (const int8_t*)buf;
    for (size_t i = 0; i < nbytes; ++i) data->push_back(Json((int64_t)v[i]));
  } else if (datatype == "UINT8") {
    for (size_t i = 0; i < nbytes; ++i)
      data->push_back(Json((int64_t)buf[i]));
  } else if (datatype == "UINT16") {
    const uint16_t* v = (const uint16_t*)buf;
    for (size_t i = 0; i < nbytes / 2; ++i)
      data->push_back(Json((int64_t)v[i]));
  } else if (datatype == "UINT32") {
    const uint32_t* v = (const uint32_t*)buf;
    for (size_t i = 0; i < nbytes / 4; ++i)
      data->push_back(Json((int64_t)v[i]));
  } else if (datatype == "UINT64") {
    const uint64_t* v = (const uint64_t*)buf;
    for (size_t i = 0; i < nbytes / 8; ++i)
      data->push_back(Json((int64_t)v[i]));
  } else if (datatype == "BOOL") {
    for (size_t i = 0; i < nbytes; ++i) data->push_back(Json(buf[i] != 0));
  } else if (datatype == "BYTES") {
    size_t pos = 0;
    while (pos + 4 <= nbytes) {
      uint32_t len;
      memcpy(&len, buf + pos, 4);
      pos += 4;
      data->push_back(
          Json(std::string((const char*)buf + pos, len)));
      pos += len;
    }
  }
  // FP16/BF16 rejected earlier (must be binary)
}

}  // namespace

namespace {

// zlib helpers for request/response bodies (reference
// http_client.cc:145-221 compresses the scatter-list the same way).
bool ZlibCompress(const std::string& in, std::string* out, bool gzip) {
  z_stream zs;
  memset(&zs, 0, sizeof(zs));
  int window = gzip ? 15 + 16 : 15;
  if (deflateInit2(&zs, Z_DEFAULT_COMPRESSION, Z_DEFLATED, window, 8,
                   Z_DEFAULT_STRATEGY) != Z_OK) {
    return false;
  }
  zs.next_in = (Bytef*)in.data();
  zs.avail_in = (uInt)in.size();
  out->resize(deflateBound(&zs, in.size()));
  zs.next_out = (Bytef*)out->data();
  zs.avail_out = (uInt)out->size();
  int rc = deflate(&zs, Z_FINISH);
  deflateEnd(&zs);
  if (rc != Z_STREAM_END) return false;
  out->resize(zs.total_out);
  return true;
}

bool ZlibDecompress(const std::string& in, std::string* out) {
  z_stream zs;
  memset(&zs, 0, sizeof(zs));
  if (inflateInit2(&zs, 15 + 32) != Z_OK) return false;  // auto gzip/zlib
  zs.next_in = (Bytef*)in.data();
  zs.avail_in = (uInt)in.size();
  out->clear();
  char buf[1 << 16];
  int rc;
  do {
    zs.next_out = (Bytef*)buf;
    zs.avail_out = sizeof(buf);
    rc = inflate(&zs, Z_NO_FLUSH);
    if (rc != Z_OK && rc != Z_STREAM_END) {
      inflateEnd(&zs);
      return false;
    }
    out->append(buf, sizeof(buf) - zs.avail_out);
  } while (rc != Z_STREAM_END && zs.avail_in > 0);
  inflateEnd(&zs);
  return rc == Z_STREAM_END;
}

}  // namespace

//==============================================================================
// InferResultHttp

void InferResultHttp::Create(
    InferResult** result, std::shared_ptr<std::string> response_body,
    size_t json_size, int http_code) {
  *result = new InferResultHttp(std::move(response_body), json_size, http_code);
}

InferResultHttp::InferResultHttp(
    std::shared_ptr<std::string> response_body, size_t json_size,
    int http_code)
    : response_body_(std::move(response_body)) {
  size_t jlen = json_size ? json_size : response_body_->size();
  try {
    response_json_ = Json::Parse(response_body_->data(), jlen);
  } catch (const std::exception& e) {
    status_ = Error(std::string("failed to parse response JSON: ") + e.what());
    return;
  }
  if (http_code == 499) {
    status_ = Error("Deadline Exceeded");
    return;
  }
  if (http_code >= 400) {
    if (response_json_.Has("error")) {
      status_ = Error(response_json_["error"].AsString());
    } else {
      status_ = Error("HTTP status " + std::to_string(http_code));
    }
    return;
  }
  binary_base_ = jlen;
  size_t offset = 0;
  if (response_json_.Has("outputs")) {
    for (const auto& out : response_json_["outputs"].AsArray()) {
      const Json& params = out["parameters"];
      if (params.Has("binary_data_size")) {
        size_t sz = (size_t)params["binary_data_size"].AsInt();
        binary_offsets_[out["name"].AsString()] = {offset, sz};
        offset += sz;
      }
    }
  }
}

const Json* InferResultHttp::FindOutput(const std::string& name) const {
  if (!response_json_.Has("outputs")) return nullptr;
  for (const auto& out : response_json_["outputs"].AsArray()) {
    if (out["name"].AsString() == name) return &out;
  }
  return nullptr;
}

Error InferResultHttp::ModelName(std::string* name) const {
  if (!response_json_.Has("model_name")) return Error("no model name");
  *name = response_json_["model_name"].AsString();
  return Error::Success;
}

Error InferResultHttp::ModelVersion(std::string* version) const {
  if (!response_json_.Has("model_version")) return Error("no model version");
  *version = response_json_["model_version"].AsString();
  return Error::Success;
}

Error InferResultHttp::Id(std::string* id) const {
  if (!response_json_.Has("id")) return Error("no id");
  *id = response_json_["id"].AsString();
  return Error::Success;
}

Error InferResultHttp::Shape(
    const std::string& output_name, std::vector<int64_t>* shape) const {
  const Json* out = FindOutput(output_name);
  if (out == nullptr)
    return Error("no result found for requested output: " + output_name);
  shape->clear();
  for (const auto& d : (*out)["shape"].AsArray()) shape->push_back(d.AsInt());
  return Error::Success;
}

Error InferResultHttp::Datatype(
    const std::string& output_name, std::string* datatype) const {
  const Json* out = FindOutput(output_name);
  if (out == nullptr)
    return Error("no result found for requested output: " + output_name);
  *datatype = (*out)["datatype"].AsString();
  return Error::Success;
}

Error InferResultHttp::RawData(
    const std::string& output_name, const uint8_t** buf,
    size_t* byte_size) const {
  auto it = binary_offsets_.find(output_name);
  if (it == binary_offsets_.end())
    return Error("no binary data found for requested output: " + output_name);
  *buf = (const uint8_t*)response_body_->data() + binary_base_ +
         it->second.first;
  *byte_size = it->second.second;
  return Error::Success;
}

//==============================================================================
// InferenceServerHttpClient

Error InferenceServerHttpClient::Create(
    std::unique_ptr<InferenceServerHttpClient>* client,
    const std::string& server_url, bool verbose) {
  client->reset(new InferenceServerHttpClient(server_url, verbose));
  return Error::Success;
}

Error InferenceServerHttpClient::Create(
    std::unique_ptr<InferenceServerHttpClient>* client,
    const std::string& server_url, bool verbose, bool use_ssl,
    const HttpSslOptions& ssl_options) {
  client->reset(new InferenceServerHttpClient(server_url, verbose));
  (*client)->use_ssl_ = use_ssl;
  (*client)->ssl_options_ = ssl_options;
  return Error::Success;
}

//==============================================================================
// TLS plumbing (sync path). OpenSSL is initialized lazily per client.

Error InferenceServerHttpClient::EnsureSslCtx() {
  // sync callers and the async worker can both get here — serialize
  // context creation (mu_ also guards new_transfers_; creation is rare)
  std::lock_guard<std::mutex> ctx_lock(mu_);
  if (ssl_ctx_ != nullptr) return Error::Success;
  static std::once_flag ssl_init;
  // SIGPIPE: SSL_write lacks MSG_NOSIGNAL; a peer reset would kill
  // the process without this (libcurl installs the same ignore).
  std::call_once(ssl_init, [] {
    SSL_library_init();
    signal(SIGPIPE, SIG_IGN);
  });
  SSL_CTX* ctx = SSL_CTX_new(TLS_client_method());
  if (ctx == nullptr) return Error("SSL_CTX_new failed");
  if (!ssl_options_.ca_info.empty()) {
    if (SSL_CTX_load_verify_locations(ctx, ssl_options_.ca_info.c_str(),
                                      nullptr) != 1) {
      SSL_CTX_free(ctx);
      return Error("failed to load CA bundle " + ssl_options_.ca_info);
    }
  } else {
    SSL_CTX_set_default_verify_paths(ctx);
  }
  if (!ssl_options_.cert.empty()) {
    if (SSL_CTX_use_certificate_file(ctx, ssl_options_.cert.c_str(),
                                     SSL_FILETYPE_PEM) != 1 ||
        SSL_CTX_use_PrivateKey_file(ctx, ssl_options_.key.c_str(),
                                    SSL_FILETYPE_PEM) != 1) {
      SSL_CTX_free(ctx);
      return Error("failed to load client cert/key");
    }
  }
  SSL_CTX_set_verify(
      ctx, ssl_options_.verify_peer ? SSL_VERIFY_PEER : SSL_VERIFY_NONE,
      nullptr);
  // async transfers retry SSL_write after partial progress
  SSL_CTX_set_mode(ctx, SSL_MODE_ENABLE_PARTIAL_WRITE |
                            SSL_MODE_ACCEPT_MOVING_WRITE_BUFFER);
  ssl_ctx_ = ctx;
  return Error::Success;
}

Error InferenceServerHttpClient::NewSsl(int fd, void** ssl_out) {
  RETURN_IF_ERROR(EnsureSslCtx());
  SSL* ssl = SSL_new((SSL_CTX*)ssl_ctx_);
  if (ssl == nullptr) return Error("SSL_new failed");
  SSL_set_fd(ssl, fd);
  SSL_set_tlsext_host_name(ssl, host_.c_str());
  if (ssl_options_.verify_host) {
    SSL_set1_host(ssl, host_.c_str());
  }
  *ssl_out = ssl;
  return Error::Success;
}

Error InferenceServerHttpClient::SslConnect() {
  void* ssl_v = nullptr;
  RETURN_IF_ERROR(NewSsl(sync_fd_, &ssl_v));
  SSL* ssl = (SSL*)ssl_v;
  if (SSL_connect(ssl) != 1) {
    unsigned long err = ERR_get_error();
    char buf[256];
    ERR_error_string_n(err, buf, sizeof(buf));
    SSL_free(ssl);
    return Error(std::string("TLS handshake failed: ") + buf);
  }
  ssl_ = ssl;
  return Error::Success;
}

void InferenceServerHttpClient::SslClose() {
  if (ssl_ != nullptr) {
    SSL_shutdown((SSL*)ssl_);
    SSL_free((SSL*)ssl_);
    ssl_ = nullptr;
  }
}

bool InferenceServerHttpClient::IoSend(const char* data, size_t n) {
  if (ssl_ != nullptr) {
    size_t sent = 0;
    while (sent < n) {
      int r = SSL_write((SSL*)ssl_, data + sent, (int)(n - sent));
      if (r <= 0) return false;
      sent += (size_t)r;
    }
    return true;
  }
  return SendAll(sync_fd_, data, n);
}

long InferenceServerHttpClient::IoRecv(char* buf, size_t n) {
  if (ssl_ != nullptr) {
    return SSL_read((SSL*)ssl_, buf, (int)n);
  }
  return recv(sync_fd_, buf, n, 0);
}

InferenceServerHttpClient::InferenceServerHttpClient(
    const std::string& url, bool verbose)
    : InferenceServerClient(verbose) {
  size_t colon = url.rfind(':');
  if (colon == std::string::npos) {
    host_ = url;
    port_ = 80;
  } else {
    host_ = url.substr(0, colon);
    port_ = atoi(url.c_str() + colon + 1);
  }
}

InferenceServerHttpClient::~InferenceServerHttpClient() {
  SslClose();
  if (ssl_ctx_ != nullptr) {
    SSL_CTX_free((SSL_CTX*)ssl_ctx_);
    ssl_ctx_ = nullptr;
  }
  exiting_ = true;
  if (worker_running_.load()) {
    // wake the worker so it can observe exiting_
    char b = 1;
    (void)!write(wakeup_fds_[1], &b, 1);
    if (worker_.joinable()) worker_.join();
  }
  if (wakeup_fds_[0] >= 0) close(wakeup_fds_[0]);
  if (wakeup_fds_[1] >= 0) close(wakeup_fds_[1]);
  if (sync_fd_ >= 0) close(sync_fd_);
}

Error InferenceServerHttpClient::DoRequest(
    int* http_code, std::string* response_body, const std::string& method,
    const std::string& path, const std::string& body, const Headers& headers,
    long timeout_us, Headers* response_headers) {
  for (int attempt = 0; attempt < 2; ++attempt) {
    if (sync_fd_ < 0) {
      sync_fd_ = ConnectTo(host_, port_, false);
      if (sync_fd_ < 0)
        return Error("failed to connect to " + host_ + ":" +
                     std::to_string(port_));
      if (use_ssl_) {
        Error ssl_err = SslConnect();
        if (!ssl_err.IsOk()) {
          close(sync_fd_);
          sync_fd_ = -1;
          return ssl_err;
        }
      }
    }
    // SO_RCVTIMEO persists on the reused fd: always (re)set it.
    struct timeval tv;
    tv.tv_sec = timeout_us / 1000000;
    tv.tv_usec = timeout_us % 1000000;
    setsockopt(sync_fd_, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    std::string req;
    req.reserve(512 + body.size());
    req += method + " " + path + " HTTP/1.1\r\n";
    req += "Host: " + host_ + ":" + std::to_string(port_) + "\r\n";
    req += "Content-Length: " + std::to_string(body.size()) + "\r\n";
    for (const auto& kv : headers) req += kv.first + ": " + kv.second + "\r\n";
    req += "\r\n";
    req += body;
    if (!IoSend(req.data(), req.size())) {
      SslClose();
      close(sync_fd_);
      sync_fd_ = -1;
      if (attempt == 1) return Error("failed to send request");
      continue;
    }
    // read response
    std::string buf;
    size_t header_end = std::string::npos;
    char chunk[kRecvChunk];
    bool conn_err = false;
    while (header_end == std::string::npos) {
      long n = IoRecv(chunk, sizeof(chunk));
      if (n <= 0) {
        if (n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK) &&
            timeout_us > 0) {
          SslClose();
          close(sync_fd_);
          sync_fd_ = -1;
          return Error("Deadline Exceeded");
        }
        conn_err = true;
        break;
      }
      buf.append(chunk, (size_t)n);
      header_end = buf.find("\r\n\r\n");
    }
    if (conn_err) {
      SslClose();
      close(sync_fd_);
      sync_fd_ = -1;
      if (attempt == 1) return Error("failed to receive response");
      continue;
    }
    Headers resp_headers;
    int code = ParseResponseHead(buf, header_end, &resp_headers);
    if (code < 0) {
      close(sync_fd_);
      sync_fd_ = -1;
      return Error("malformed HTTP response");
    }
    size_t body_start = header_end + 4;
    size_t content_length = 0;
    auto it = resp_headers.find("content-length");
    if (it != resp_headers.end()) content_length = (size_t)atoll(it->second.c_str());
    while (buf.size() - body_start < content_length) {
      long n = IoRecv(chunk, sizeof(chunk));
      if (n <= 0) {
        bool timed_out = n < 0 && (errno == EAGAIN || errno == EWOULDBLOCK) &&
                         timeout_us > 0;
        SslClose();
        close(sync_fd_);
        sync_fd_ = -1;
        return Error(
            timed_out ? "Deadline Exceeded" : "connection closed mid-body");
      }
      buf.append(chunk, (size_t)n);
    }
    *http_code = code;
    response_body->assign(buf, body_start, content_length);
    if (response_headers != nullptr) *response_headers = resp_headers;
    return Error::Success;
  }
  return Error("unreachable");
}

Error InferenceServerHttpClient::Get(
    std::string* response, const std::string& path, const Headers& headers,
    bool* ok_flag) {
  int code;
  RETURN_IF_ERROR(DoRequest(&code, response, "GET", path, "", headers));
  if (ok_flag != nullptr) {
    *ok_flag = (code == 200);
    return Error::Success;
  }
  if (code >= 400) {
    try {
      Json err = Json::Parse(*response);
      if (err.Has("error")) return Error(err["error"].AsString());
    } catch (...) {
    }
    return Error("HTTP status " + std::to_string(code));
  }
  return Error::Success;
}

Error InferenceServerHttpClient::Post(
    std::string* response, const std::string& path, const std::string& body,
    const Headers& headers) {
  int code;
  RETURN_IF_ERROR(DoRequest(&code, response, "POST", path, body, headers));
  if (code >= 400) {
    try {
      Json err = Json::Parse(*response);
      if (err.Has("error")) return Error(err["error"].AsString());
    } catch (...) {
    }
    return Error("HTTP status " + std::to_string(code));
  }
  return Error::Success;
}

// ---- health / metadata ----

Error InferenceServerHttpClient::IsServerLive(
    bool* live, const Headers& headers) {
  std::string r;
  return Get(&r, "/v2/health/live", headers, live);
}

Error InferenceServerHttpClient::IsServerReady(
    bool* ready, const Headers& headers) {
  std::string r;
  return Get(&r, "/v2/health/ready", headers, ready);
}

Error InferenceServerHttpClient::IsModelReady(
    bool* ready, const std::string& model_name,
    const std::string& model_version, const Headers& headers) {
  std::string path = "/v2/models/" + UrlEncode(model_name);
  if (!model_version.empty()) path += "/versions/" + model_version;
  path += "/ready";
  std::string r;
  return Get(&r, path, headers, ready);
}

Error InferenceServerHttpClient::ServerMetadata(
    std::string* server_metadata, const Headers& headers) {
  return Get(server_metadata, "/v2", headers);
}

Error InferenceServerHttpClient::ModelMetadata(
    std::string* model_metadata, const std::string& model_name,
    const std::string& model_version, const Headers& headers) {
  std::string path = "/v2/models/" + UrlEncode(model_name);
  if (!model_version.empty()) path += "/versions/" + model_version;
  return Get(model_metadata, path, headers);
}

Error InferenceServerHttpClient::ModelConfig(
    std::string* model_config, const std::string& model_name,
    const std::string& model_version, const Headers& headers) {
  std::string path = "/v2/models/" + UrlEncode(model_name);
  if (!model_version.empty()) path += "/versions/" + model_version;
  path += "/config";
  return Get(model_config, path, headers);
}

// ---- repository ----

Error InferenceServerHttpClient::ModelRepositoryIndex(
    std::string* repository_index, const Headers& headers) {
  return Post(repository_index, "/v2/repository/index", "", headers);
}

Error InferenceServerHttpClient::LoadModel(
    const std::string& model_name, const Headers& headers,
    const std::string& config,
    const std::map<std::string, std::vector<char>>& files) {
  Json req;
  JsonObject params;
  if (!config.empty()) params["config"] = Json(config);
  for (const auto& kv : files) {
    params[kv.first] = Json(Base64Encode(
        (const uint8_t*)kv.second.data(), kv.second.size()));
  }
  if (!params.empty()) req.Set("parameters", Json(std::move(params)));
  std::string r;
  return Post(
      &r, "/v2/repository/models/" + UrlEncode(model_name) + "/load",
      req.IsNull() ? "{}" : req.Dump(), headers);
}

Error InferenceServerHttpClient::UnloadModel(
    const std::string& model_name, const Headers& headers) {
  std::string r;
  return Post(
      &r, "/v2/repository/models/" + UrlEncode(model_name) + "/unload", "{}",
      headers);
}

// ---- statistics / trace / logging ----

Error InferenceServerHttpClient::ModelInferenceStatistics(
    std::string* infer_stat, const std::string& model_name,
    const std::string& model_version, const Headers& headers) {
  std::string path;
  if (!model_name.empty()) {
    path = "/v2/models/" + UrlEncode(model_name);
    if (!model_version.empty()) path += "/versions/" + model_version;
    path += "/stats";
  } else {
    path = "/v2/models/stats";
  }
  return Get(infer_stat, path, headers);
}

Error InferenceServerHttpClient::UpdateTraceSettings(
    std::string* response, const std::string& model_name,
    const std::map<std::string, std::vector<std::string>>& settings,
    const Headers& headers) {
  Json req;
  for (const auto& kv : settings) {
    if (kv.second.empty()) {
      req.Set(kv.first, Json());
    } else if (kv.second.size() == 1) {
      req.Set(kv.first, Json(kv.second[0]));
    } else {
      JsonArray arr;
      for (const auto& v : kv.second) arr.push_back(Json(v));
      req.Set(kv.first, Json(std::move(arr)));
    }
  }
  std::string path = model_name.empty()
                         ? "/v2/trace/setting"
                         : "/v2/models/" + UrlEncode(model_name) +
                               "/trace/setting";
  return Post(response, path, req.IsNull() ? "{}" : req.Dump(), headers);
}

Error InferenceServerHttpClient::GetTraceSettings(
    std::string* settings, const std::string& model_name,
    const Headers& headers) {
  std::string path = model_name.empty()
                         ? "/v2/trace/setting"
                         : "/v2/models/" + UrlEncode(model_name) +
                               "/trace/setting";
  return Get(settings, path, headers);
}

Error InferenceServerHttpClient::UpdateLogSettings(
    std::string* response, const std::map<std::string, std::string>& settings,
    const Headers& headers) {
  Json req;
  for (const auto& kv : settings) req.Set(kv.first, Json(kv.second));
  return Post(response, "/v2/logging", req.IsNull() ? "{}" : req.Dump(),
              headers);
}

Error InferenceServerHttpClient::GetLogSettings(
    std::string* settings, const Headers& headers) {
  return Get(settings, "/v2/logging", headers);
}

// ---- shared memory ----

Error InferenceServerHttpClient::SystemSharedMemoryStatus(
    std::string* status, const std::string& region_name,
    const Headers& headers) {
  std::string path =
      region_name.empty()
          ? "/v2/systemsharedmemory/status"
          : "/v2/systemsharedmemory/region/" + UrlEncode(region_name) +
                "/status";
  return Get(status, path, headers);
}

Error InferenceServerHttpClient::RegisterSystemSharedMemory(
    const std::string& name, const std::string& key, size_t byte_size,
    size_t offset, const Headers& headers) {
  Json req;
  req.Set("key", Json(key));
  req.Set("offset", Json((int64_t)offset));
  req.Set("byte_size", Json((int64_t)byte_size));
  std::string r;
  return Post(
      &r, "/v2/systemsharedmemory/region/" + UrlEncode(name) + "/register",
      req.Dump(), headers);
}

Error InferenceServerHttpClient::UnregisterSystemSharedMemory(
    const std::string& name, const Headers& headers) {
  std::string path = name.empty()
                         ? "/v2/systemsharedmemory/unregister"
                         : "/v2/systemsharedmemory/region/" + UrlEncode(name) +
                               "/unregister";
  std::string r;
  return Post(&r, path, "", headers);
}

Error InferenceServerHttpClient::CudaSharedMemoryStatus(
    std::string* status, const std::string& region_name,
    const Headers& headers) {
  std::string path =
      region_name.empty()
          ? "/v2/cudasharedmemory/status"
          : "/v2/cudasharedmemory/region/" + UrlEncode(region_name) +
                "/status";
  return Get(status, path, headers);
}

Error InferenceServerHttpClient::RegisterCudaSharedMemory(
    const std::string& name, const std::string& raw_handle, size_t device_id,
    size_t byte_size, const Headers& headers) {
  Json req;
  Json b64;
  b64.Set("b64", Json(Base64Encode(raw_handle)));
  req.Set("raw_handle", std::move(b64));
  req.Set("device_id", Json((int64_t)device_id));
  req.Set("byte_size", Json((int64_t)byte_size));
  std::string r;
  return Post(
      &r, "/v2/cudasharedmemory/region/" + UrlEncode(name) + "/register",
      req.Dump(), headers);
}

Error InferenceServerHttpClient::UnregisterCudaSharedMemory(
    const std::string& name, const Headers& headers) {
  std::string path = name.empty()
                         ? "/v2/cudasharedmemory/unregister"
                         : "/v2/cudasharedmemory/region/" + UrlEncode(name) +
                               "/unregister";
  std::string r;
  return Post(&r, path, "", headers);
}

// ---- inference ----

Error InferenceServerHttpClient::GenerateRequestBody(
    std::vector<char>* request_body, size_t* header_length,
    const InferOptions& options, const std::vector<InferInput*>& inputs,
    const std::vector<const InferRequestedOutput*>& outputs) {
  Json req;
  if (!options.request_id_.empty()) req.Set("id", Json(options.request_id_));
  JsonObject params;
  if (options.sequence_id_ != 0 || !options.sequence_id_str_.empty()) {
    if (!options.sequence_id_str_.empty()) {
      params["sequence_id"] = Json(options.sequence_id_str_);
    } else {
      params["sequence_id"] = Json((int64_t)options.sequence_id_);
    }
    params["sequence_start"] = Json(options.sequence_start_);
    params["sequence_end"] = Json(options.sequence_end_);
  }
  if (options.priority_ != 0)
    params["priority"] = Json((int64_t)options.priority_);
  if (options.server_timeout_ != 0)
    params["timeout"] = Json((int64_t)options.server_timeout_);
  for (const auto& kv : options.request_parameters_)
    params[kv.first] = Json(kv.second);

  JsonArray json_inputs;
  std::vector<std::pair<const uint8_t*, size_t>> binary_segments;
  size_t total_binary = 0;
  for (InferInput* input : inputs) {
    Json jin;
    jin.Set("name", Json(input->Name()));
    JsonArray shape;
    for (int64_t d : input->Shape()) shape.push_back(Json(d));
    jin.Set("shape", Json(std::move(shape)));
    jin.Set("datatype", Json(input->Datatype()));
    JsonObject in_params;
    if (input->IsSharedMemory()) {
      std::string region;
      size_t size, offset;
      input->SharedMemoryInfo(&region, &size, &offset);
      in_params["shared_memory_region"] = Json(region);
      in_params["shared_memory_byte_size"] = Json((int64_t)size);
      if (offset != 0)
        in_params["shared_memory_offset"] = Json((int64_t)offset);
    } else if (input->BinaryData()) {
      in_params["binary_data_size"] = Json((int64_t)input->ByteSize());
      input->PrepareForRequest();
      const uint8_t* buf;
      size_t n;
      bool end = false;
      while (!end) {
        input->GetNext(&buf, &n, &end);
        if (buf != nullptr && n > 0) {
          binary_segments.emplace_back(buf, n);
          total_binary += n;
        }
      }
    } else {
      if (input->Datatype() == "FP16" || input->Datatype() == "BF16") {
        return Error(
            input->Datatype() +
            " inputs must use binary data over HTTP");
      }
      // JSON data path: concatenate segments then convert
      input->PrepareForRequest();
      std::string contiguous;
      const uint8_t* buf;
      size_t n;
      bool end = false;
      while (!end) {
        input->GetNext(&buf, &n, &end);
        if (buf != nullptr && n > 0)
          contiguous.append((const char*)buf, n);
      }
      JsonArray data;
      AppendDataJson(&data, input->Datatype(),
                     (const uint8_t*)contiguous.data(), contiguous.size());
      jin.Set("data", Json(std::move(data)));
    }
    if (!in_params.empty()) jin.Set("parameters", Json(std::move(in_params)));
    json_inputs.push_back(std::move(jin));
  }
  req.Set("inputs", Json(std::move(json_inputs)));

  if (!outputs.empty()) {
    JsonArray json_outputs;
    for (const InferRequestedOutput* output : outputs) {
      Json jout;
      jout.Set("name", Json(output->Name()));
      JsonObject out_params;
      if (output->IsSharedMemory()) {
        std::string region;
        size_t size, offset;
        output->SharedMemoryInfo(&region, &size, &offset);
        out_params["shared_memory_region"] = Json(region);
        out_params["shared_memory_byte_size"] = Json((int64_t)size);
        if (offset != 0)
          out_params["shared_memory_offset"] = Json((int64_t)offset);
      } else {
        out_params["binary_data"] = Json(output->BinaryData());
      }
      if (output->ClassCount() != 0)
        out_params["classification"] = Json((int64_t)output->ClassCount());
      if (!out_params.empty())
        jout.Set("parameters", Json(std::move(out_params)));
      json_outputs.push_back(std::move(jout));
    }
    req.Set("outputs", Json(std::move(json_outputs)));
  } else {
    params["binary_data_output"] = Json(true);
  }
  if (!params.empty()) req.Set("parameters", Json(std::move(params)));

  std::string json_str = req.Dump();
  *header_length = json_str.size();
  request_body->clear();
  request_body->reserve(json_str.size() + total_binary);
  request_body->insert(request_body->end(), json_str.begin(), json_str.end());
  for (const auto& seg : binary_segments) {
    request_body->insert(
        request_body->end(), seg.first, seg.first + seg.second);
  }
  return Error::Success;
}

Error InferenceServerHttpClient::ParseResponseBody(
    InferResult** result, const std::vector<char>& response_body,
    size_t header_length) {
  auto body = std::make_shared<std::string>(
      response_body.data(), response_body.size());
  InferResultHttp::Create(result, body, header_length, 200);
  return Error::Success;
}

Error InferenceServerHttpClient::Infer(
    InferResult** result, const InferOptions& options,
    const std::vector<InferInput*>& inputs,
    const std::vector<const InferRequestedOutput*>& outputs,
    const Headers& headers, const Parameters& query_params,
    const CompressionType request_compression_algorithm,
    const CompressionType response_compression_algorithm) {
  RequestTimers timer;
  timer.CaptureTimestamp(RequestTimers::Kind::REQUEST_START);

  std::vector<char> body;
  size_t header_length;
  RETURN_IF_ERROR(
      GenerateRequestBody(&body, &header_length, options, inputs, outputs));

  Headers hdrs = headers;
  if (header_length != body.size()) {
    hdrs["Inference-Header-Content-Length"] = std::to_string(header_length);
  }
  hdrs["Content-Type"] = "application/octet-stream";
  std::string compressed;
  if (request_compression_algorithm != CompressionType::NONE) {
    bool gzip = request_compression_algorithm == CompressionType::GZIP;
    std::string raw(body.data(), body.size());
    if (!ZlibCompress(raw, &compressed, gzip)) {
      return Error("failed to compress request body");
    }
    body.assign(compressed.begin(), compressed.end());
    hdrs["Content-Encoding"] = gzip ? "gzip" : "deflate";
  }
  if (response_compression_algorithm == CompressionType::GZIP) {
    hdrs["Accept-Encoding"] = "gzip";
  } else if (response_compression_algorithm == CompressionType::DEFLATE) {
    hdrs["Accept-Encoding"] = "deflate";
  }

  std::string path = "/v2/models/" + UrlEncode(options.model_name_);
  if (!options.model_version_.empty())
    path += "/versions/" + options.model_version_;
  path += "/infer";
  if (!query_params.empty()) {
    path += "?";
    bool first = true;
    for (const auto& kv : query_params) {
      if (!first) path += "&";
      first = false;
      path += UrlEncode(kv.first) + "=" + UrlEncode(kv.second);
    }
  }

  timer.CaptureTimestamp(RequestTimers::Kind::SEND_START);
  int http_code;
  std::string response;
  Headers response_headers;
  Error err = DoRequest(
      &http_code, &response, "POST", path,
      std::string(body.data(), body.size()), hdrs,
      (long)options.client_timeout_, &response_headers);
  timer.CaptureTimestamp(RequestTimers::Kind::SEND_END);
  timer.CaptureTimestamp(RequestTimers::Kind::RECV_START);
  if (!err.IsOk()) return err;
  timer.CaptureTimestamp(RequestTimers::Kind::RECV_END);

  size_t json_size = 0;
  auto it = response_headers.find("inference-header-content-length");
  if (it != response_headers.end()) json_size = (size_t)atoll(it->second.c_str());
  auto enc = response_headers.find("content-encoding");
  if (enc != response_headers.end()) {
    std::string decompressed;
    if (!ZlibDecompress(response, &decompressed)) {
      return Error("failed to decompress response body");
    }
    response = std::move(decompressed);
  }

  auto body_ptr = std::make_shared<std::string>(std::move(response));
  InferResultHttp::Create(result, body_ptr, json_size, http_code);

  timer.CaptureTimestamp(RequestTimers::Kind::REQUEST_END);
  UpdateInferStat(timer);
  return (*result)->RequestStatus();
}

Error InferenceServerHttpClient::InferMulti(
    std::vector<InferResult*>* results, const std::vector<InferOptions>& options,
    const std::vector<std::vector<InferInput*>>& inputs,
    const std::vector<std::vector<const InferRequestedOutput*>>& outputs,
    const Headers& headers, const Parameters& query_params) {
  // options/outputs may have size 1 (shared) or match inputs 1:1
  // (reference http_client.cc:1912-1955)
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
    Error err = Infer(&result, opt, inputs[i], outs, headers, query_params);
    results->push_back(result);
    if (!err.IsOk()) return err;
  }
  return Error::Success;
}

//==============================================================================
// Async path: epoll worker multiplexing non-blocking transfers (the
// curl-multi analog, reference http_client.cc:2249-2348).

struct InferenceServerHttpClient::AsyncTransfer {
  int fd = -1;
  SSL* ssl = nullptr;      // per-transfer TLS (worker thread only)
  bool tls_ready = false;  // handshake finished
  std::string out;     // full request bytes
  size_t out_pos = 0;
  std::string in;      // accumulated response
  size_t header_end = std::string::npos;
  size_t content_length = 0;
  size_t body_start = 0;
  bool headers_parsed = false;
  int http_code = 0;
  Headers response_headers;
  OnCompleteFn callback;
  RequestTimers timer;
};

Error InferenceServerHttpClient::AsyncInfer(
    OnCompleteFn callback, const InferOptions& options,
    const std::vector<InferInput*>& inputs,
    const std::vector<const InferRequestedOutput*>& outputs,
    const Headers& headers, const Parameters& query_params) {
  if (callback == nullptr)
    return Error("callback must not be null for AsyncInfer");

  auto transfer = std::make_unique<AsyncTransfer>();
  transfer->timer.CaptureTimestamp(RequestTimers::Kind::REQUEST_START);

  std::vector<char> body;
  size_t header_length;
  RETURN_IF_ERROR(
      GenerateRequestBody(&body, &header_length, options, inputs, outputs));

  std::string path = "/v2/models/" + UrlEncode(options.model_name_);
  if (!options.model_version_.empty())
    path += "/versions/" + options.model_version_;
  path += "/infer";

  std::string& req = transfer->out;
  req += "POST " + path + " HTTP/1.1\r\n";
  req += "Host: " + host_ + ":" + std::to_string(port_) + "\r\n";
  req += "Content-Length: " + std::to_string(body.size()) + "\r\n";
  req += "Content-Type: application/octet-stream\r\n";
  if (header_length != body.size()) {
    req += "Inference-Header-Content-Length: " +
           std::to_string(header_length) + "\r\n";
  }
  for (const auto& kv : headers) req += kv.first + ": " + kv.second + "\r\n";
  req += "\r\n";
  req.append(body.data(), body.size());
  transfer->callback = std::move(callback);

  {
    std::lock_guard<std::mutex> lock(mu_);
    if (!worker_running_.load()) {
      if (pipe(wakeup_fds_) != 0) return Error("failed to create wakeup pipe");
      fcntl(wakeup_fds_[0], F_SETFL, O_NONBLOCK);
      worker_running_ = true;
      worker_ = std::thread(&InferenceServerHttpClient::AsyncWorker, this);
    }
    new_transfers_.push_back(std::move(transfer));
  }
  char b = 1;
  (void)!write(wakeup_fds_[1], &b, 1);
  return Error::Success;
}

Error InferenceServerHttpClient::AsyncInferMulti(
    OnMultiCompleteFn callback, const std::vector<InferOptions>& options,
    const std::vector<std::vector<InferInput*>>& inputs,
    const std::vector<std::vector<const InferRequestedOutput*>>& outputs,
    const Headers& headers, const Parameters& query_params) {
  if (options.size() != 1 && options.size() != inputs.size()) {
    return Error("'options' must be of size 1 or match 'inputs'");
  }
  if (!outputs.empty() && outputs.size() != 1 &&
      outputs.size() != inputs.size()) {
    return Error("'outputs' must be empty, size 1, or match 'inputs'");
  }
  // atomic countdown join -> single callback (reference :1984-2003)
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
        opt, inputs[i], outs, headers, query_params);
    if (!err.IsOk()) return err;
  }
  return Error::Success;
}

void InferenceServerHttpClient::AsyncWorker() {
  int epfd = epoll_create1(0);
  struct epoll_event ev;
  ev.events = EPOLLIN;
  ev.data.ptr = nullptr;  // wakeup marker
  epoll_ctl(epfd, EPOLL_CTL_ADD, wakeup_fds_[0], &ev);

  std::map<int, std::unique_ptr<AsyncTransfer>> active;

  auto finish = [&](std::unique_ptr<AsyncTransfer> t, int code,
                    const std::string& error_msg) {
    if (t->ssl != nullptr) {
      SSL_free(t->ssl);
      t->ssl = nullptr;
    }
    if (t->fd >= 0) {
      epoll_ctl(epfd, EPOLL_CTL_DEL, t->fd, nullptr);
      close(t->fd);
    }
    t->timer.CaptureTimestamp(RequestTimers::Kind::RECV_END);
    t->timer.CaptureTimestamp(RequestTimers::Kind::REQUEST_END);
    InferResult* result = nullptr;
    if (!error_msg.empty()) {
      auto body = std::make_shared<std::string>(
          "{\"error\":\"" + error_msg + "\"}");
      InferResultHttp::Create(&result, body, 0, 400);
    } else {
      size_t json_size = 0;
      auto it = t->response_headers.find("inference-header-content-length");
      if (it != t->response_headers.end())
        json_size = (size_t)atoll(it->second.c_str());
      auto body = std::make_shared<std::string>(
          t->in.substr(t->body_start, t->content_length));
      InferResultHttp::Create(&result, body, json_size, code);
    }
    UpdateInferStat(t->timer);
    t->callback(result);
  };

  char chunk[kRecvChunk];
  while (!exiting_) {
    // adopt new transfers: drain the queue under mu_, then do the
    // slow work (connect, TLS object setup) without holding it so
    // AsyncInfer submitters never block behind a connect()
    std::deque<std::unique_ptr<AsyncTransfer>> adopted;
    {
      std::lock_guard<std::mutex> lock(mu_);
      adopted.swap(new_transfers_);
    }
    for (auto& tp : adopted) {
      auto t = std::move(tp);
      t->fd = ConnectTo(host_, port_, true);
      if (t->fd < 0) {
        finish(std::move(t), 0, "failed to connect");
        continue;
      }
      if (use_ssl_) {
        void* ssl_v = nullptr;
        Error serr = NewSsl(t->fd, &ssl_v);
        if (!serr.IsOk()) {
          finish(std::move(t), 0, serr.Message());
          continue;
        }
        t->ssl = (SSL*)ssl_v;
      }
      t->timer.CaptureTimestamp(RequestTimers::Kind::SEND_START);
      struct epoll_event tev;
      tev.events = EPOLLOUT | EPOLLIN;
      tev.data.fd = t->fd;
      epoll_ctl(epfd, EPOLL_CTL_ADD, t->fd, &tev);
      active[t->fd] = std::move(t);
    }

    struct epoll_event events[64];
    int n = epoll_wait(epfd, events, 64, 100);
    for (int i = 0; i < n; ++i) {
      if (events[i].data.ptr == nullptr && events[i].data.fd == 0) {
        // ambiguous zero fd marker — handled below via wakeup read
      }
      if (events[i].data.ptr == nullptr) {
        // wakeup pipe
        while (read(wakeup_fds_[0], chunk, sizeof(chunk)) > 0) {
        }
        continue;
      }
      int fd = events[i].data.fd;
      auto it = active.find(fd);
      if (it == active.end()) continue;
      AsyncTransfer* t = it->second.get();

      if (events[i].events & (EPOLLERR | EPOLLHUP)) {
        auto owned = std::move(it->second);
        active.erase(it);
        finish(std::move(owned), 0, "connection error");
        continue;
      }
      auto set_events = [&](uint32_t mask) {
        struct epoll_event tev;
        tev.events = mask;
        tev.data.fd = fd;
        epoll_ctl(epfd, EPOLL_CTL_MOD, fd, &tev);
      };
      // Non-blocking TLS handshake woven into the event machine: the
      // first SSL_connect also absorbs TCP connect-in-progress (write
      // returns EAGAIN until the connect completes -> WANT_WRITE).
      if (t->ssl != nullptr && !t->tls_ready) {
        int hr = SSL_connect(t->ssl);
        if (hr == 1) {
          t->tls_ready = true;
          set_events(EPOLLOUT | EPOLLIN);
        } else {
          int e = SSL_get_error(t->ssl, hr);
          if (e == SSL_ERROR_WANT_READ) {
            set_events(EPOLLIN);  // avoid level-triggered EPOLLOUT spin
            continue;
          } else if (e == SSL_ERROR_WANT_WRITE) {
            set_events(EPOLLOUT | EPOLLIN);
            continue;
          } else {
            unsigned long ee = ERR_get_error();
            char ebuf[256];
            ERR_error_string_n(ee, ebuf, sizeof(ebuf));
            auto owned = std::move(it->second);
            active.erase(it);
            finish(std::move(owned), 0,
                   std::string("TLS handshake failed: ") + ebuf);
            continue;
          }
        }
      }
      if ((events[i].events & EPOLLOUT) && t->out_pos < t->out.size()) {
        while (t->out_pos < t->out.size()) {
          ssize_t sent;
          if (t->ssl != nullptr) {
            int r = SSL_write(t->ssl, t->out.data() + t->out_pos,
                              (int)(t->out.size() - t->out_pos));
            if (r > 0) {
              sent = r;
            } else {
              int e = SSL_get_error(t->ssl, r);
              sent = -1;
              errno = (e == SSL_ERROR_WANT_WRITE || e == SSL_ERROR_WANT_READ)
                          ? EAGAIN
                          : EPIPE;
            }
          } else {
            sent = send(fd, t->out.data() + t->out_pos,
                        t->out.size() - t->out_pos, MSG_NOSIGNAL);
          }
          if (sent > 0) {
            t->out_pos += (size_t)sent;
          } else if (sent < 0 && (errno == EAGAIN || errno == EWOULDBLOCK)) {
            break;
          } else {
            break;
          }
        }
        if (t->out_pos >= t->out.size()) {
          t->timer.CaptureTimestamp(RequestTimers::Kind::SEND_END);
          set_events(EPOLLIN);
        }
      }
      if (events[i].events & EPOLLIN) {
        bool closed = false;
        while (true) {
          ssize_t r;
          if (t->ssl != nullptr) {
            int sr = SSL_read(t->ssl, chunk, sizeof(chunk));
            if (sr > 0) {
              r = sr;
            } else {
              int e = SSL_get_error(t->ssl, sr);
              if (e == SSL_ERROR_WANT_READ || e == SSL_ERROR_WANT_WRITE) {
                r = -1;
                errno = EAGAIN;
              } else {
                r = 0;  // clean close or fatal -> treat as closed
              }
            }
          } else {
            r = recv(fd, chunk, sizeof(chunk), 0);
          }
          if (r > 0) {
            if (t->in.empty())
              t->timer.CaptureTimestamp(RequestTimers::Kind::RECV_START);
            t->in.append(chunk, (size_t)r);
          } else if (r == 0) {
            closed = true;
            break;
          } else if (errno == EAGAIN || errno == EWOULDBLOCK) {
            break;
          } else {
            closed = true;
            break;
          }
        }
        if (!t->headers_parsed) {
          t->header_end = t->in.find("\r\n\r\n");
          if (t->header_end != std::string::npos) {
            t->http_code =
                ParseResponseHead(t->in, t->header_end, &t->response_headers);
            auto cl = t->response_headers.find("content-length");
            t->content_length =
                cl != t->response_headers.end()
                    ? (size_t)atoll(cl->second.c_str())
                    : 0;
            t->body_start = t->header_end + 4;
            t->headers_parsed = true;
          }
        }
        if (t->headers_parsed &&
            t->in.size() - t->body_start >= t->content_length) {
          int code = t->http_code;
          auto owned = std::move(it->second);
          active.erase(it);
          finish(std::move(owned), code, "");
          continue;
        }
        if (closed) {
          auto owned = std::move(it->second);
          active.erase(it);
          finish(std::move(owned), 0, "connection closed");
        }
      }
    }
  }
  // drain remaining transfers with error
  for (auto& kv : active) {
    finish(std::move(kv.second), 0, "client shutting down");
  }
  close(epfd);
}

}  // namespace client_amd
