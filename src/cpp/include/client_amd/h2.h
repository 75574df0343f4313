// HTTP/2 (h2c, RFC 7540) client connection — written from scratch for
// the gRPC transport: framing, SETTINGS, PING, flow control (connection
// + stream windows, both directions), HEADERS/CONTINUATION, and HPACK
// encoding (literal, static-table references). HPACK *decoding* uses
// the system libnghttp2 inflater (the same library the reference links
// via libcurl; ABI declared locally — no dev headers in this image).
//
// One reader thread per connection dispatches frames to per-stream
// handlers; all writes serialize through a mutex. Streams are created
// in monotonically increasing odd IDs as RFC 7540 §5.1.1 requires.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <functional>
#include <map>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "client_amd/common.h"

namespace client_amd {

using HeaderList = std::vector<std::pair<std::string, std::string>>;

// TLS settings for an h2 connection (gRPC "grpcs"). Fields hold PEM
// *contents* (not paths), matching the reference grpc_client.h
// SslOptions:112-136 semantics where callers read files themselves.
struct H2SslOptions {
  bool use_ssl = false;
  std::string root_certificates;  // PEM CA bundle; empty = system default
  std::string private_key;        // client key PEM (mTLS)
  std::string certificate_chain;  // client cert PEM (mTLS)
};

// h2 PING keepalive (gRPC keepalive.md semantics): send a PING every
// time_ms; if the ACK doesn't arrive within timeout_ms the connection
// is declared dead and every in-flight stream fails.
struct H2KeepAlive {
  bool enabled = false;
  int time_ms = 0;
  int timeout_ms = 20000;
  bool permit_without_calls = false;
  int max_pings_without_data = 2;  // 0 = unlimited
};

class H2Connection {
 public:
  struct StreamHandler {
    std::function<void(const HeaderList&)> on_headers;
    std::function<void(const uint8_t*, size_t)> on_data;
    // final headers (trailers) or END_STREAM; stream is finished after
    std::function<void(const HeaderList&)> on_trailers;
    std::function<void(const std::string&)> on_error;
  };

  H2Connection() = default;
  ~H2Connection();

  Error Connect(const std::string& host, int port);
  Error Connect(const std::string& host, int port, const H2SslOptions& ssl);
  // Start the PING watchdog (call once, after Connect succeeds).
  void StartKeepalive(const H2KeepAlive& ka);
  void Close();
  bool IsAlive() const { return alive_; }

  // Send HEADERS (no END_STREAM) opening a new stream; returns id.
  Error StartStream(const HeaderList& headers, StreamHandler handler,
                    int32_t* stream_id);
  // Send DATA, chunked to the peer's max frame size, blocking on flow
  // control windows. end_stream closes our half.
  Error SendData(int32_t stream_id, const uint8_t* data, size_t n,
                 bool end_stream);
  // Half-close our side with an empty DATA frame.
  Error FinishStream(int32_t stream_id);
  // RST_STREAM (cancel).
  void ResetStream(int32_t stream_id, uint32_t error_code = 0x8 /*CANCEL*/);

 private:
  struct Stream {
    StreamHandler handler;
    bool saw_headers = false;
    bool closed = false;
    int64_t send_window = 65535;
    int64_t recv_consumed = 0;
  };

  void ReaderLoop();
  void KeepaliveLoop();
  bool WriteRaw(const uint8_t* data, size_t n);
  Error TlsHandshake(const std::string& host, const H2SslOptions& ssl);
  // Blocking-semantics read: >0 bytes, <=0 on close/error. Plain mode is
  // a blocking recv(); TLS mode polls a non-blocking fd and serializes
  // SSL_read/SSL_write through io_mu_ (one SSL object, two threads).
  ssize_t IoRecv(char* buf, size_t n);
  bool WriteFrame(uint8_t type, uint8_t flags, int32_t stream_id,
                  const std::string& payload);
  void HandleFrame(uint8_t type, uint8_t flags, int32_t stream_id,
                   const uint8_t* payload, size_t len);
  void FailAllStreams(const std::string& msg);
  std::string EncodeHeaders(const HeaderList& headers);

  int fd_ = -1;
  bool use_tls_ = false;
  void* ssl_ctx_ = nullptr;  // SSL_CTX* (avoid OpenSSL headers here)
  void* ssl_ = nullptr;      // SSL*
  std::mutex io_mu_;         // guards every SSL_read/SSL_write
  std::thread reader_;
  std::atomic_bool alive_{false};
  std::atomic_bool exiting_{false};  // also read outside mu_

  std::mutex write_mu_;
  int32_t next_stream_id_ = 1;

  std::mutex mu_;  // protects streams_ + windows
  std::condition_variable window_cv_;
  std::map<int32_t, Stream> streams_;
  int64_t conn_send_window_ = 65535;
  int64_t conn_recv_consumed_ = 0;
  uint32_t peer_max_frame_ = 16384;
  int32_t peer_initial_window_ = 65535;

  // keepalive watchdog state (guarded by mu_; thread woken via ka_cv_)
  H2KeepAlive ka_;
  std::thread keepalive_;
  std::condition_variable ka_cv_;
  uint64_t pings_acked_ = 0;
  uint64_t data_epoch_ = 0;        // bumped on every HEADERS/DATA we send
  uint64_t last_ping_epoch_ = 0;   // data_epoch_ when the last ping went out
  int pings_without_data_ = 0;

  // HEADERS/CONTINUATION reassembly
  int32_t pending_headers_stream_ = 0;
  uint8_t pending_headers_flags_ = 0;
  std::string pending_headers_block_;

  void* hpack_inflater_ = nullptr;  // nghttp2_hd_inflater*
};

}  // namespace client_amd
