#include "client_amd/h2.h"

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <openssl/err.h>
#include <openssl/pem.h>
#include <openssl/ssl.h>
#include <openssl/x509v3.h>
#include <poll.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <signal.h>

#include <atomic>
#include <mutex>

// ---------------------------------------------------------------------------
// libnghttp2 HPACK inflater ABI (system libnghttp2.so.14; prototypes
// declared here because the image ships no dev headers — the struct and
// functions below are the long-stable public C ABI of nghttp2 ≥1.x).
extern "C" {
typedef struct nghttp2_hd_inflater nghttp2_hd_inflater;
typedef struct {
  uint8_t* name;
  uint8_t* value;
  size_t namelen;
  size_t valuelen;
  uint8_t flags;
} nghttp2_nv;
int nghttp2_hd_inflate_new(nghttp2_hd_inflater** inflater_ptr);
void nghttp2_hd_inflate_del(nghttp2_hd_inflater* inflater);
long nghttp2_hd_inflate_hd2(nghttp2_hd_inflater* inflater, nghttp2_nv* nv_out,
                            int* inflate_flags, const uint8_t* in,
                            size_t inlen, int in_final);
int nghttp2_hd_inflate_end_headers(nghttp2_hd_inflater* inflater);
}
#define NGHTTP2_HD_INFLATE_FINAL 0x01
#define NGHTTP2_HD_INFLATE_EMIT 0x02

namespace client_amd {

namespace {

constexpr uint8_t kFrameData = 0x0;
constexpr uint8_t kFrameHeaders = 0x1;
constexpr uint8_t kFrameRstStream = 0x3;
constexpr uint8_t kFrameSettings = 0x4;
constexpr uint8_t kFramePing = 0x6;
constexpr uint8_t kFrameGoaway = 0x7;
constexpr uint8_t kFrameWindowUpdate = 0x8;
constexpr uint8_t kFrameContinuation = 0x9;

constexpr uint8_t kFlagEndStream = 0x1;
constexpr uint8_t kFlagEndHeaders = 0x4;
constexpr uint8_t kFlagAck = 0x1;
constexpr uint8_t kFlagPadded = 0x8;
constexpr uint8_t kFlagPriority = 0x20;

constexpr int64_t kRecvWindow = 1 << 28;       // advertised via SETTINGS
constexpr int64_t kRecvReplenish = 1 << 24;    // WINDOW_UPDATE threshold

void be24(std::string* out, uint32_t v) {
  out->push_back((char)((v >> 16) & 0xFF));
  out->push_back((char)((v >> 8) & 0xFF));
  out->push_back((char)(v & 0xFF));
}

void be32(std::string* out, uint32_t v) {
  out->push_back((char)((v >> 24) & 0xFF));
  out->push_back((char)((v >> 16) & 0xFF));
  out->push_back((char)((v >> 8) & 0xFF));
  out->push_back((char)(v & 0xFF));
}

uint32_t rd32(const uint8_t* p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
         ((uint32_t)p[2] << 8) | p[3];
}

}  // namespace

H2Connection::~H2Connection() { Close(); }

Error H2Connection::Connect(const std::string& host, int port) {
  return Connect(host, port, H2SslOptions());
}

// Load a PEM bundle (possibly several certs) into the context's trust
// store. Reference grpc clients take PEM strings, not paths, so we read
// from a memory BIO rather than SSL_CTX_load_verify_locations.
static Error AddPemRoots(SSL_CTX* ctx, const std::string& pem) {
  BIO* bio = BIO_new_mem_buf(pem.data(), (int)pem.size());
  if (bio == nullptr) return Error("BIO_new_mem_buf failed");
  X509_STORE* store = SSL_CTX_get_cert_store(ctx);
  int added = 0;
  while (X509* cert = PEM_read_bio_X509(bio, nullptr, nullptr, nullptr)) {
    X509_STORE_add_cert(store, cert);
    X509_free(cert);
    added++;
  }
  BIO_free(bio);
  ERR_clear_error();  // trailing PEM_R_NO_START_LINE from the read loop
  if (added == 0) return Error("no certificates found in root PEM");
  return Error::Success;
}

Error H2Connection::TlsHandshake(const std::string& host,
                                 const H2SslOptions& ssl_opts) {
  // SSL_write has no MSG_NOSIGNAL: a peer-closed socket would raise
  // SIGPIPE and kill the process. Same global-ignore libcurl installs.
  static std::once_flag sigpipe_once;
  std::call_once(sigpipe_once, [] { signal(SIGPIPE, SIG_IGN); });
  SSL_CTX* ctx = SSL_CTX_new(TLS_client_method());
  if (ctx == nullptr) return Error("SSL_CTX_new failed");
  ssl_ctx_ = ctx;
  if (!ssl_opts.root_certificates.empty()) {
    Error err = AddPemRoots(ctx, ssl_opts.root_certificates);
    if (!err.IsOk()) return err;
  } else {
    SSL_CTX_set_default_verify_paths(ctx);
  }
  if (!ssl_opts.certificate_chain.empty()) {
    BIO* cbio = BIO_new_mem_buf(ssl_opts.certificate_chain.data(),
                                (int)ssl_opts.certificate_chain.size());
    X509* cert =
        cbio ? PEM_read_bio_X509(cbio, nullptr, nullptr, nullptr) : nullptr;
    if (cbio) BIO_free(cbio);
    BIO* kbio = BIO_new_mem_buf(ssl_opts.private_key.data(),
                                (int)ssl_opts.private_key.size());
    EVP_PKEY* key =
        kbio ? PEM_read_bio_PrivateKey(kbio, nullptr, nullptr, nullptr)
             : nullptr;
    if (kbio) BIO_free(kbio);
    bool ok = cert != nullptr && key != nullptr &&
              SSL_CTX_use_certificate(ctx, cert) == 1 &&
              SSL_CTX_use_PrivateKey(ctx, key) == 1;
    if (cert) X509_free(cert);
    if (key) EVP_PKEY_free(key);
    if (!ok) return Error("failed to load client certificate/key PEM");
  }
  SSL_CTX_set_verify(ctx, SSL_VERIFY_PEER, nullptr);
  // gRPC requires ALPN "h2" (RFC 7301); grpcio rejects TLS without it.
  static const unsigned char kAlpn[] = {2, 'h', '2'};
  SSL_CTX_set_alpn_protos(ctx, kAlpn, sizeof(kAlpn));

  SSL* ssl = SSL_new(ctx);
  if (ssl == nullptr) return Error("SSL_new failed");
  ssl_ = ssl;
  SSL_set_fd(ssl, fd_);
  SSL_set_mode(ssl, SSL_MODE_ENABLE_PARTIAL_WRITE |
                        SSL_MODE_ACCEPT_MOVING_WRITE_BUFFER);
  // Hostname checks: IP literals match iPAddress SANs, names match
  // dNSName SANs (+SNI).
  unsigned char ipbuf[16];
  if (inet_pton(AF_INET, host.c_str(), ipbuf) == 1) {
    X509_VERIFY_PARAM_set1_ip(SSL_get0_param(ssl), ipbuf, 4);
  } else if (inet_pton(AF_INET6, host.c_str(), ipbuf) == 1) {
    X509_VERIFY_PARAM_set1_ip(SSL_get0_param(ssl), ipbuf, 16);
  } else {
    SSL_set_tlsext_host_name(ssl, host.c_str());
    SSL_set1_host(ssl, host.c_str());
  }
  if (SSL_connect(ssl) != 1) {
    unsigned long e = ERR_get_error();
    char ebuf[256];
    ERR_error_string_n(e, ebuf, sizeof(ebuf));
    long vr = SSL_get_verify_result(ssl);
    std::string msg = std::string("TLS handshake with ") + host +
                      " failed: " + ebuf;
    if (vr != X509_V_OK) {
      msg += std::string(" (verify: ") +
             X509_verify_cert_error_string(vr) + ")";
    }
    return Error(msg);
  }
  const unsigned char* alpn = nullptr;
  unsigned int alpn_len = 0;
  SSL_get0_alpn_selected(ssl, &alpn, &alpn_len);
  if (alpn != nullptr && !(alpn_len == 2 && memcmp(alpn, "h2", 2) == 0)) {
    return Error("server negotiated ALPN protocol other than h2");
  }
  // Steady state runs the socket non-blocking so SSL_read/SSL_write
  // never block while holding io_mu_ (reader + writer threads share ssl_).
  int fl = fcntl(fd_, F_GETFL, 0);
  fcntl(fd_, F_SETFL, fl | O_NONBLOCK);
  use_tls_ = true;
  return Error::Success;
}

Error H2Connection::Connect(const std::string& host, int port,
                            const H2SslOptions& ssl_opts) {
  struct addrinfo hints;
  memset(&hints, 0, sizeof(hints));
  hints.ai_family = AF_UNSPEC;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  if (getaddrinfo(host.c_str(), std::to_string(port).c_str(), &hints, &res) !=
      0) {
    return Error("failed to resolve " + host);
  }
  for (struct addrinfo* rp = res; rp != nullptr; rp = rp->ai_next) {
    fd_ = socket(rp->ai_family, rp->ai_socktype, rp->ai_protocol);
    if (fd_ < 0) continue;
    int one = 1;
    setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    if (connect(fd_, rp->ai_addr, rp->ai_addrlen) == 0) break;
    close(fd_);
    fd_ = -1;
  }
  freeaddrinfo(res);
  if (fd_ < 0) {
    return Error("failed to connect to " + host + ":" + std::to_string(port));
  }

  if (ssl_opts.use_ssl) {
    Error err = TlsHandshake(host, ssl_opts);
    if (!err.IsOk()) {
      if (ssl_ != nullptr) SSL_free((SSL*)ssl_);
      if (ssl_ctx_ != nullptr) SSL_CTX_free((SSL_CTX*)ssl_ctx_);
      ssl_ = nullptr;
      ssl_ctx_ = nullptr;
      close(fd_);
      fd_ = -1;
      return err;
    }
  }

  nghttp2_hd_inflater* inf = nullptr;
  if (nghttp2_hd_inflate_new(&inf) != 0) {
    return Error("failed to create HPACK inflater");
  }
  hpack_inflater_ = inf;

  // client preface + SETTINGS(ENABLE_PUSH=0, INITIAL_WINDOW_SIZE) +
  // connection WINDOW_UPDATE to grow the 65535 default
  static const char kPreface[] = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
  std::string settings;
  auto setting = [&](uint16_t id, uint32_t v) {
    settings.push_back((char)(id >> 8));
    settings.push_back((char)(id & 0xFF));
    be32(&settings, v);
  };
  setting(0x2, 0);                     // ENABLE_PUSH = 0
  setting(0x4, (uint32_t)kRecvWindow); // INITIAL_WINDOW_SIZE
  setting(0x6, 1 << 20);               // MAX_HEADER_LIST_SIZE

  std::string buf(kPreface, sizeof(kPreface) - 1);
  be24(&buf, (uint32_t)settings.size());
  buf.push_back((char)kFrameSettings);
  buf.push_back(0);
  be32(&buf, 0);
  buf += settings;
  // connection window update
  be24(&buf, 4);
  buf.push_back((char)kFrameWindowUpdate);
  buf.push_back(0);
  be32(&buf, 0);
  be32(&buf, (uint32_t)(kRecvWindow - 65535));
  if (!WriteRaw((const uint8_t*)buf.data(), buf.size())) {
    return Error("failed to send HTTP/2 preface");
  }
  alive_ = true;
  reader_ = std::thread(&H2Connection::ReaderLoop, this);
  return Error::Success;
}

void H2Connection::Close() {
  {
    std::lock_guard<std::mutex> lock(mu_);
    if (exiting_) return;
    exiting_ = true;
  }
  alive_ = false;
  // shutdown() BEFORE joining: the keepalive thread may be blocked in a
  // socket send (WriteFrame) and the reader in recv — both need the fd
  // torn down to return.
  if (fd_ >= 0) {
    shutdown(fd_, SHUT_RDWR);
  }
  ka_cv_.notify_all();
  if (keepalive_.joinable()) keepalive_.join();
  if (reader_.joinable()) reader_.join();
  if (ssl_ != nullptr) {
    SSL_free((SSL*)ssl_);
    ssl_ = nullptr;
  }
  if (ssl_ctx_ != nullptr) {
    SSL_CTX_free((SSL_CTX*)ssl_ctx_);
    ssl_ctx_ = nullptr;
  }
  if (fd_ >= 0) {
    close(fd_);
    fd_ = -1;
  }
  if (hpack_inflater_ != nullptr) {
    nghttp2_hd_inflate_del((nghttp2_hd_inflater*)hpack_inflater_);
    hpack_inflater_ = nullptr;
  }
  window_cv_.notify_all();
}

void H2Connection::StartKeepalive(const H2KeepAlive& ka) {
  if (!ka.enabled || ka.time_ms <= 0 || keepalive_.joinable()) return;
  {
    std::lock_guard<std::mutex> lock(mu_);
    ka_ = ka;
  }
  keepalive_ = std::thread(&H2Connection::KeepaliveLoop, this);
}

// gRPC-style transport watchdog (grpc/doc/keepalive.md): PING every
// time_ms; a missing ACK within timeout_ms kills the connection so
// blocked callers fail fast instead of hanging on a dead peer.
void H2Connection::KeepaliveLoop() {
  std::unique_lock<std::mutex> lock(mu_);
  while (!exiting_ && alive_) {
    ka_cv_.wait_for(lock, std::chrono::milliseconds(ka_.time_ms));
    if (exiting_ || !alive_) return;
    if (!ka_.permit_without_calls && streams_.empty()) continue;
    if (data_epoch_ == last_ping_epoch_) {
      if (ka_.max_pings_without_data > 0 &&
          pings_without_data_ >= ka_.max_pings_without_data) {
        continue;  // idle transport: stop pinging (grpc-core rule)
      }
      pings_without_data_++;
    } else {
      pings_without_data_ = 0;
    }
    last_ping_epoch_ = data_epoch_;
    uint64_t acked_before = pings_acked_;
    lock.unlock();
    bool sent = WriteFrame(kFramePing, 0, 0, std::string(8, '\0'));
    lock.lock();
    bool acked =
        sent && ka_cv_.wait_for(
                    lock, std::chrono::milliseconds(ka_.timeout_ms), [&] {
                      return exiting_ || pings_acked_ > acked_before;
                    });
    if (exiting_) return;
    if (!acked) {
      lock.unlock();
      alive_ = false;
      if (fd_ >= 0) shutdown(fd_, SHUT_RDWR);
      FailAllStreams("keepalive watchdog timed out");
      return;
    }
  }
}

bool H2Connection::WriteRaw(const uint8_t* data, size_t n) {
  if (!use_tls_) {
    size_t sent = 0;
    while (sent < n) {
      ssize_t r = send(fd_, data + sent, n - sent, MSG_NOSIGNAL);
      if (r <= 0) {
        if (r < 0 && errno == EINTR) continue;
        return false;
      }
      sent += (size_t)r;
    }
    return true;
  }
  // TLS: non-blocking fd; retry WANT_READ/WANT_WRITE via poll. io_mu_ is
  // held across the poll — the reader's SSL_read waits, which is the same
  // backpressure a blocking send() gives the plain path.
  std::lock_guard<std::mutex> iolock(io_mu_);
  SSL* ssl = (SSL*)ssl_;
  size_t sent = 0;
  while (sent < n) {
    int r = SSL_write(ssl, data + sent, (int)(n - sent));
    if (r > 0) {
      sent += (size_t)r;
      continue;
    }
    int e = SSL_get_error(ssl, r);
    if (e != SSL_ERROR_WANT_READ && e != SSL_ERROR_WANT_WRITE) return false;
    struct pollfd pfd = {fd_, (short)(e == SSL_ERROR_WANT_READ ? POLLIN
                                                               : POLLOUT),
                         0};
    if (poll(&pfd, 1, 5000) < 0 && errno != EINTR) return false;
    if (pfd.revents & (POLLERR | POLLNVAL)) return false;
  }
  return true;
}

ssize_t H2Connection::IoRecv(char* buf, size_t n) {
  if (!use_tls_) return recv(fd_, buf, n, 0);
  while (true) {
    {
      std::lock_guard<std::mutex> iolock(io_mu_);
      SSL* ssl = (SSL*)ssl_;
      int r = SSL_read(ssl, buf, (int)n);
      if (r > 0) return r;
      int e = SSL_get_error(ssl, r);
      if (e != SSL_ERROR_WANT_READ && e != SSL_ERROR_WANT_WRITE) {
        return 0;  // clean close, fatal TLS error, or fd shut down
      }
    }
    // Wait (lock released so writers can make progress) for the socket;
    // Close()'s shutdown() wakes this poll and the next SSL_read fails.
    struct pollfd pfd = {fd_, POLLIN, 0};
    if (poll(&pfd, 1, 200) < 0 && errno != EINTR) return 0;
    if (pfd.revents & (POLLERR | POLLNVAL)) return 0;
  }
}

bool H2Connection::WriteFrame(uint8_t type, uint8_t flags, int32_t stream_id,
                              const std::string& payload) {
  std::string buf;
  be24(&buf, (uint32_t)payload.size());
  buf.push_back((char)type);
  buf.push_back((char)flags);
  be32(&buf, (uint32_t)stream_id);
  buf += payload;
  std::lock_guard<std::mutex> lock(write_mu_);
  return WriteRaw((const uint8_t*)buf.data(), buf.size());
}

// HPACK encoding: every header as "literal without indexing — new
// name" (RFC 7541 §6.2.2, prefix 0000). Valid against any decoder; no
// dynamic-table state to keep in sync.
std::string H2Connection::EncodeHeaders(const HeaderList& headers) {
  std::string out;
  auto put_len = [&](size_t len) {
    // 7-bit prefix integer with H=0
    if (len < 127) {
      out.push_back((char)len);
    } else {
      out.push_back((char)127);
      size_t rest = len - 127;
      while (rest >= 128) {
        out.push_back((char)(0x80 | (rest & 0x7F)));
        rest >>= 7;
      }
      out.push_back((char)rest);
    }
  };
  for (const auto& kv : headers) {
    out.push_back('\0');  // literal w/o indexing, new name
    put_len(kv.first.size());
    out += kv.first;
    put_len(kv.second.size());
    out += kv.second;
  }
  return out;
}

Error H2Connection::StartStream(const HeaderList& headers,
                                StreamHandler handler, int32_t* stream_id) {
  if (!alive_) return Error("h2 connection is closed");
  std::string block = EncodeHeaders(headers);
  std::lock_guard<std::mutex> wlock(write_mu_);
  int32_t id = next_stream_id_;
  next_stream_id_ += 2;
  {
    std::lock_guard<std::mutex> lock(mu_);
    Stream& s = streams_[id];
    s.handler = std::move(handler);
    s.send_window = peer_initial_window_;
    data_epoch_++;
  }
  std::string buf;
  be24(&buf, (uint32_t)block.size());
  buf.push_back((char)kFrameHeaders);
  buf.push_back((char)kFlagEndHeaders);
  be32(&buf, (uint32_t)id);
  buf += block;
  if (!WriteRaw((const uint8_t*)buf.data(), buf.size())) {
    return Error("failed to send HEADERS");
  }
  *stream_id = id;
  return Error::Success;
}

Error H2Connection::SendData(int32_t stream_id, const uint8_t* data, size_t n,
                             bool end_stream) {
  {
    std::lock_guard<std::mutex> lock(mu_);
    data_epoch_++;
    if (streams_.find(stream_id) == streams_.end()) {
      // stream already finished/reset: sending DATA would be a
      // protocol violation that could kill the (shared) connection
      return Error("stream is closed");
    }
  }
  size_t pos = 0;
  while (pos < n || (end_stream && n == 0)) {
    size_t want = n - pos;
    {
      std::unique_lock<std::mutex> lock(mu_);
      window_cv_.wait(lock, [&] {
        return exiting_ ||
               (conn_send_window_ > 0 &&
                streams_[stream_id].send_window > 0) ||
               want == 0;
      });
      if (exiting_) return Error("h2 connection closed");
      if (want > 0) {
        int64_t allowed = std::min<int64_t>(
            {(int64_t)want, conn_send_window_,
             streams_[stream_id].send_window, (int64_t)peer_max_frame_});
        want = (size_t)allowed;
        conn_send_window_ -= allowed;
        streams_[stream_id].send_window -= allowed;
      }
    }
    bool last = (pos + want >= n);
    std::string payload((const char*)data + pos, want);
    if (!WriteFrame(kFrameData, (last && end_stream) ? kFlagEndStream : 0,
                    stream_id, payload)) {
      return Error("failed to send DATA");
    }
    pos += want;
    if (last) break;
  }
  return Error::Success;
}

Error H2Connection::FinishStream(int32_t stream_id) {
  {
    std::lock_guard<std::mutex> lock(mu_);
    if (streams_.find(stream_id) == streams_.end()) {
      return Error::Success;  // already finished server-side
    }
  }
  if (!WriteFrame(kFrameData, kFlagEndStream, stream_id, "")) {
    return Error("failed to half-close stream");
  }
  return Error::Success;
}

void H2Connection::ResetStream(int32_t stream_id, uint32_t error_code) {
  std::string payload;
  be32(&payload, error_code);
  WriteFrame(kFrameRstStream, 0, stream_id, payload);
  std::lock_guard<std::mutex> lock(mu_);
  streams_.erase(stream_id);
}

void H2Connection::FailAllStreams(const std::string& msg) {
  std::map<int32_t, Stream> streams;
  {
    std::lock_guard<std::mutex> lock(mu_);
    streams.swap(streams_);
  }
  for (auto& kv : streams) {
    if (kv.second.handler.on_error) kv.second.handler.on_error(msg);
  }
  window_cv_.notify_all();
}

void H2Connection::ReaderLoop() {
  std::string buf;
  char chunk[1 << 16];
  while (true) {
    // need 9-byte frame header
    while (buf.size() < 9) {
      ssize_t r = IoRecv(chunk, sizeof(chunk));
      if (r <= 0) {
        alive_ = false;
        FailAllStreams("h2 connection closed by peer");
        return;
      }
      buf.append(chunk, (size_t)r);
    }
    uint32_t len = ((uint32_t)(uint8_t)buf[0] << 16) |
                   ((uint32_t)(uint8_t)buf[1] << 8) | (uint8_t)buf[2];
    uint8_t type = (uint8_t)buf[3];
    uint8_t flags = (uint8_t)buf[4];
    int32_t stream_id = (int32_t)(rd32((const uint8_t*)buf.data() + 5) &
                                  0x7FFFFFFF);
    while (buf.size() < 9 + len) {
      ssize_t r = IoRecv(chunk, sizeof(chunk));
      if (r <= 0) {
        alive_ = false;
        FailAllStreams("h2 connection closed mid-frame");
        return;
      }
      buf.append(chunk, (size_t)r);
    }
    HandleFrame(type, flags, stream_id, (const uint8_t*)buf.data() + 9, len);
    buf.erase(0, 9 + len);
    {
      std::lock_guard<std::mutex> lock(mu_);
      if (exiting_) return;
    }
  }
}

void H2Connection::HandleFrame(uint8_t type, uint8_t flags, int32_t stream_id,
                               const uint8_t* payload, size_t len) {
  switch (type) {
    case kFrameData: {
      if (flags & kFlagPadded) {
        uint8_t pad = payload[0];
        payload += 1;
        len = len >= 1u + pad ? len - 1 - pad : 0;
      }
      StreamHandler handler;
      bool have = false;
      bool end = (flags & kFlagEndStream) != 0;
      {
        std::lock_guard<std::mutex> lock(mu_);
        auto it = streams_.find(stream_id);
        if (it != streams_.end()) {
          handler = it->second.handler;
          have = true;
          it->second.recv_consumed += (int64_t)len;
          conn_recv_consumed_ += (int64_t)len;
        }
      }
      if (have && len > 0 && handler.on_data) handler.on_data(payload, len);
      // replenish receive windows
      int64_t conn_replenish = 0;
      int64_t stream_replenish = 0;
      {
        std::lock_guard<std::mutex> lock(mu_);
        if (conn_recv_consumed_ >= kRecvReplenish) {
          conn_replenish = conn_recv_consumed_;
          conn_recv_consumed_ = 0;
        }
        auto it = streams_.find(stream_id);
        if (it != streams_.end() && it->second.recv_consumed >= kRecvReplenish
            && !end) {
          stream_replenish = it->second.recv_consumed;
          it->second.recv_consumed = 0;
        }
      }
      if (conn_replenish > 0) {
        std::string p;
        be32(&p, (uint32_t)conn_replenish);
        WriteFrame(kFrameWindowUpdate, 0, 0, p);
      }
      if (stream_replenish > 0) {
        std::string p;
        be32(&p, (uint32_t)stream_replenish);
        WriteFrame(kFrameWindowUpdate, 0, stream_id, p);
      }
      if (have && end) {
        {
          std::lock_guard<std::mutex> lock(mu_);
          streams_.erase(stream_id);
        }
        if (handler.on_trailers) handler.on_trailers({});
      }
      break;
    }
    case kFrameHeaders:
    case kFrameContinuation: {
      const uint8_t* block = payload;
      size_t block_len = len;
      if (type == kFrameHeaders) {
        pending_headers_stream_ = stream_id;
        pending_headers_flags_ = flags;
        pending_headers_block_.clear();
        if (flags & kFlagPadded) {
          uint8_t pad = block[0];
          block += 1;
          block_len = block_len >= 1u + pad ? block_len - 1 - pad : 0;
        }
        if (flags & kFlagPriority) {
          block += 5;
          block_len = block_len >= 5 ? block_len - 5 : 0;
        }
      }
      pending_headers_block_.append((const char*)block, block_len);
      bool end_headers = (type == kFrameHeaders)
                             ? (flags & kFlagEndHeaders)
                             : (flags & kFlagEndHeaders);
      if (!end_headers) break;

      // decode accumulated block
      HeaderList decoded;
      auto* inf = (nghttp2_hd_inflater*)hpack_inflater_;
      const uint8_t* in = (const uint8_t*)pending_headers_block_.data();
      size_t in_len = pending_headers_block_.size();
      while (true) {
        nghttp2_nv nv;
        int inflate_flags = 0;
        long rv = nghttp2_hd_inflate_hd2(inf, &nv, &inflate_flags, in, in_len,
                                         1);
        if (rv < 0) {
          FailAllStreams("HPACK decode error");
          return;
        }
        in += rv;
        in_len -= (size_t)rv;
        if (inflate_flags & NGHTTP2_HD_INFLATE_EMIT) {
          decoded.emplace_back(
              std::string((const char*)nv.name, nv.namelen),
              std::string((const char*)nv.value, nv.valuelen));
        }
        if (inflate_flags & NGHTTP2_HD_INFLATE_FINAL) {
          nghttp2_hd_inflate_end_headers(inf);
          break;
        }
        if (in_len == 0 && !(inflate_flags & NGHTTP2_HD_INFLATE_EMIT)) {
          break;
        }
      }

      int32_t sid = pending_headers_stream_;
      bool end_stream = (pending_headers_flags_ & kFlagEndStream) != 0;
      StreamHandler handler;
      bool have = false;
      bool is_trailers = false;
      {
        std::lock_guard<std::mutex> lock(mu_);
        auto it = streams_.find(sid);
        if (it != streams_.end()) {
          handler = it->second.handler;
          have = true;
          is_trailers = it->second.saw_headers;
          it->second.saw_headers = true;
          if (end_stream) streams_.erase(it);
        }
      }
      if (have) {
        if (!is_trailers && handler.on_headers) handler.on_headers(decoded);
        if (end_stream && handler.on_trailers) {
          handler.on_trailers(is_trailers ? decoded : HeaderList{});
        }
      }
      break;
    }
    case kFrameSettings: {
      if (flags & kFlagAck) break;
      {
        std::lock_guard<std::mutex> lock(mu_);
        for (size_t i = 0; i + 6 <= len; i += 6) {
          uint16_t id = ((uint16_t)payload[i] << 8) | payload[i + 1];
          uint32_t value = rd32(payload + i + 2);
          if (id == 0x4) {  // INITIAL_WINDOW_SIZE
            int64_t delta = (int64_t)value - peer_initial_window_;
            peer_initial_window_ = (int32_t)value;
            for (auto& kv : streams_) kv.second.send_window += delta;
          } else if (id == 0x5) {  // MAX_FRAME_SIZE
            peer_max_frame_ = value;
          }
        }
      }
      window_cv_.notify_all();
      WriteFrame(kFrameSettings, kFlagAck, 0, "");
      break;
    }
    case kFramePing: {
      if (!(flags & kFlagAck)) {
        WriteFrame(kFramePing, kFlagAck, 0,
                   std::string((const char*)payload, len));
      } else {
        std::lock_guard<std::mutex> lock(mu_);
        pings_acked_++;
        ka_cv_.notify_all();
      }
      break;
    }
    case kFrameWindowUpdate: {
      uint32_t increment = rd32(payload) & 0x7FFFFFFF;
      {
        std::lock_guard<std::mutex> lock(mu_);
        if (stream_id == 0) {
          conn_send_window_ += increment;
        } else {
          auto it = streams_.find(stream_id);
          if (it != streams_.end()) it->second.send_window += increment;
        }
      }
      window_cv_.notify_all();
      break;
    }
    case kFrameRstStream: {
      uint32_t code = rd32(payload);
      StreamHandler handler;
      bool have = false;
      {
        std::lock_guard<std::mutex> lock(mu_);
        auto it = streams_.find(stream_id);
        if (it != streams_.end()) {
          handler = it->second.handler;
          have = true;
          streams_.erase(it);
        }
      }
      if (have && handler.on_error) {
        handler.on_error("stream reset by server (code " +
                         std::to_string(code) + ")");
      }
      break;
    }
    case kFrameGoaway: {
      alive_ = false;
      FailAllStreams("server sent GOAWAY");
      break;
    }
    default:
      break;  // ignore unknown frames (PRIORITY etc.)
  }
}

}  // namespace client_amd
