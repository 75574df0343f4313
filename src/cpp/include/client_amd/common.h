// Shared client API surface — Error / InferOptions / InferInput /
// InferRequestedOutput / InferResult / RequestTimers / InferStat.
// API-compatible with the reference's src/c++/library/common.h (same
// class and method names; e.g. Error :61-83, InferStat :93-114,
// InferOptions :164-231, InferInput :237-394, RequestTimers :568-648)
// but written from scratch for the MI355X stack: no CUDA types anywhere
// — the GPU surface is hip_shm.h.
#pragma once

#include <atomic>
#include <chrono>
#include <cstdint>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

namespace client_amd {

//==============================================================================
// Error status returned by client methods (value type).
//
class Error {
 public:
  explicit Error(const std::string& msg = "") : msg_(msg) {}
  const std::string& Message() const { return msg_; }
  bool IsOk() const { return msg_.empty(); }
  static const Error Success;

 private:
  std::string msg_;
  friend std::ostream& operator<<(std::ostream&, const Error&);
};

#define RETURN_IF_ERROR(S)            \
  do {                                \
    const client_amd::Error& e = (S); \
    if (!e.IsOk()) return e;          \
  } while (false)

//==============================================================================
// Per-request nanosecond timestamps (reference common.h:568-648). gRPC
// uses SEND as marshal time; HTTP sets SEND/RECV from socket events.
//
class RequestTimers {
 public:
  enum class Kind {
    REQUEST_START,
    REQUEST_END,
    SEND_START,
    SEND_END,
    RECV_START,
    RECV_END,
    COUNT_
  };

  RequestTimers() { Reset(); }
  void Reset() {
    for (auto& t : timestamps_) t = 0;
  }
  void CaptureTimestamp(Kind kind) {
    timestamps_[(size_t)kind] =
        std::chrono::duration_cast<std::chrono::nanoseconds>(
            std::chrono::steady_clock::now().time_since_epoch())
            .count();
  }
  uint64_t Timestamp(Kind kind) const { return timestamps_[(size_t)kind]; }
  uint64_t Duration(Kind start, Kind end) const {
    uint64_t s = Timestamp(start), e = Timestamp(end);
    if (s == 0 || e == 0 || e < s) return UINT64_MAX;  // invalid sentinel
    return e - s;
  }

 private:
  uint64_t timestamps_[(size_t)Kind::COUNT_];
};

//==============================================================================
// Cumulative client-side statistics (reference common.h:93-114).
//
struct InferStat {
  size_t completed_request_count = 0;
  uint64_t cumulative_total_request_time_ns = 0;
  uint64_t cumulative_send_time_ns = 0;
  uint64_t cumulative_receive_time_ns = 0;
};

//==============================================================================
// Request options (reference common.h:164-231).
//
struct InferOptions {
  explicit InferOptions(const std::string& model_name)
      : model_name_(model_name) {}

  std::string model_name_;
  std::string model_version_;
  std::string request_id_;
  uint64_t sequence_id_ = 0;
  std::string sequence_id_str_;
  bool sequence_start_ = false;
  bool sequence_end_ = false;
  uint64_t priority_ = 0;
  uint64_t server_timeout_ = 0;        // microseconds, 0 = server default
  uint64_t client_timeout_ = 0;        // microseconds, 0 = no timeout
  bool triton_enable_empty_final_response_ = false;
  std::map<std::string, std::string> request_parameters_;
};

//==============================================================================
// One named input tensor. Zero-copy: AppendRaw() records (ptr,size)
// segments; GetNext() exposes a cursor for streaming upload (reference
// common.h:237-394). SetSharedMemory() carries only region params.
//
class InferInput {
 public:
  static Error Create(
      InferInput** infer_input, const std::string& name,
      const std::vector<int64_t>& dims, const std::string& datatype);

  const std::string& Name() const { return name_; }
  const std::string& Datatype() const { return datatype_; }
  const std::vector<int64_t>& Shape() const { return shape_; }
  Error SetShape(const std::vector<int64_t>& dims) {
    shape_ = dims;
    return Error::Success;
  }

  Error Reset() {
    bufs_.clear();
    buf_byte_sizes_.clear();
    str_bufs_.clear();
    bufs_idx_ = 0;
    buf_pos_ = 0;
    byte_size_ = 0;
    io_type_ = NONE;
    return Error::Success;
  }

  Error AppendRaw(const uint8_t* input, size_t input_byte_size) {
    bufs_.push_back(input);
    buf_byte_sizes_.push_back(input_byte_size);
    byte_size_ += input_byte_size;
    io_type_ = RAW;
    return Error::Success;
  }
  Error AppendRaw(const std::vector<uint8_t>& input) {
    return AppendRaw(input.data(), input.size());
  }

  // BYTES tensor from strings: 4-byte LE length prefix per element.
  Error AppendFromString(const std::vector<std::string>& input);

  Error SetSharedMemory(
      const std::string& region_name, size_t byte_size, size_t offset = 0) {
    shm_name_ = region_name;
    shm_byte_size_ = byte_size;
    shm_offset_ = offset;
    io_type_ = SHARED_MEMORY;
    return Error::Success;
  }

  // binary vs JSON transport of this input over HTTP
  Error SetBinaryData(bool binary_data) {
    binary_data_ = binary_data;
    return Error::Success;
  }
  bool BinaryData() const { return binary_data_; }

  bool IsSharedMemory() const { return io_type_ == SHARED_MEMORY; }
  Error SharedMemoryInfo(
      std::string* name, size_t* byte_size, size_t* offset) const {
    if (io_type_ != SHARED_MEMORY) return Error("not shared memory");
    *name = shm_name_;
    *byte_size = shm_byte_size_;
    *offset = shm_offset_;
    return Error::Success;
  }

  size_t ByteSize() const { return byte_size_; }

  // streaming-upload cursor (reference common.h:274-293)
  Error PrepareForRequest() {
    bufs_idx_ = 0;
    buf_pos_ = 0;
    return Error::Success;
  }
  Error GetNext(
      const uint8_t** buf, size_t* input_bytes, bool* end_of_input) {
    if (bufs_idx_ < bufs_.size()) {
      *buf = bufs_[bufs_idx_];
      *input_bytes = buf_byte_sizes_[bufs_idx_];
      bufs_idx_++;
    } else {
      *buf = nullptr;
      *input_bytes = 0;
    }
    *end_of_input = (bufs_idx_ >= bufs_.size());
    return Error::Success;
  }

 private:
  InferInput(
      const std::string& name, const std::vector<int64_t>& dims,
      const std::string& datatype)
      : name_(name), shape_(dims), datatype_(datatype) {}

  enum IOType { NONE, RAW, SHARED_MEMORY };
  std::string name_;
  std::vector<int64_t> shape_;
  std::string datatype_;
  IOType io_type_ = NONE;
  std::vector<const uint8_t*> bufs_;
  std::vector<size_t> buf_byte_sizes_;
  std::vector<std::string> str_bufs_;  // owns AppendFromString storage
  size_t bufs_idx_ = 0;
  size_t buf_pos_ = 0;
  size_t byte_size_ = 0;
  bool binary_data_ = true;
  std::string shm_name_;
  size_t shm_byte_size_ = 0;
  size_t shm_offset_ = 0;
};

//==============================================================================
// Requested output (reference common.h:400-482).
//
class InferRequestedOutput {
 public:
  static Error Create(
      InferRequestedOutput** infer_output, const std::string& name,
      const size_t class_count = 0);

  const std::string& Name() const { return name_; }
  size_t ClassCount() const { return class_count_; }

  Error SetSharedMemory(
      const std::string& region_name, size_t byte_size, size_t offset = 0) {
    shm_name_ = region_name;
    shm_byte_size_ = byte_size;
    shm_offset_ = offset;
    is_shm_ = true;
    return Error::Success;
  }
  Error UnsetSharedMemory() {
    is_shm_ = false;
    return Error::Success;
  }
  bool IsSharedMemory() const { return is_shm_; }
  Error SharedMemoryInfo(
      std::string* name, size_t* byte_size, size_t* offset) const {
    if (!is_shm_) return Error("not shared memory");
    *name = shm_name_;
    *byte_size = shm_byte_size_;
    *offset = shm_offset_;
    return Error::Success;
  }
  Error SetBinaryData(bool binary_data) {
    binary_data_ = binary_data;
    return Error::Success;
  }
  bool BinaryData() const { return binary_data_; }

 private:
  InferRequestedOutput(const std::string& name, size_t class_count)
      : name_(name), class_count_(class_count) {}
  std::string name_;
  size_t class_count_;
  bool is_shm_ = false;
  bool binary_data_ = true;
  std::string shm_name_;
  size_t shm_byte_size_ = 0;
  size_t shm_offset_ = 0;
};

//==============================================================================
// Abstract inference result (reference common.h:488-563).
//
class InferResult {
 public:
  virtual ~InferResult() = default;
  virtual Error ModelName(std::string* name) const = 0;
  virtual Error ModelVersion(std::string* version) const = 0;
  virtual Error Id(std::string* id) const = 0;
  virtual Error Shape(
      const std::string& output_name, std::vector<int64_t>* shape) const = 0;
  virtual Error Datatype(
      const std::string& output_name, std::string* datatype) const = 0;
  virtual Error RawData(
      const std::string& output_name, const uint8_t** buf,
      size_t* byte_size) const = 0;
  virtual Error StringData(
      const std::string& output_name,
      std::vector<std::string>* string_result) const;
  virtual std::string DebugString() const = 0;
  virtual Error RequestStatus() const = 0;
};

using OnCompleteFn = std::function<void(InferResult*)>;
using OnMultiCompleteFn = std::function<void(std::vector<InferResult*>)>;

//==============================================================================
// Client base: stat accumulation + async worker scaffold (reference
// common.h:119-153, common.cc:55-).
//
class InferenceServerClient {
 public:
  explicit InferenceServerClient(bool verbose)
      : verbose_(verbose), exiting_(false) {}
  virtual ~InferenceServerClient() = default;

  Error ClientInferStat(InferStat* infer_stat) const {
    std::lock_guard<std::mutex> lock(stat_mu_);
    *infer_stat = infer_stat_;
    return Error::Success;
  }

 protected:
  // called from worker threads AND caller threads (async completions)
  Error UpdateInferStat(const RequestTimers& timer);

  bool verbose_;
  // read lock-free by worker loops (while (!exiting_)), written by the
  // destructor thread — must be atomic (TSAN-verified)
  std::atomic<bool> exiting_;
  mutable std::mutex stat_mu_;
  InferStat infer_stat_;
};

}  // namespace client_amd
