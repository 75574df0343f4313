#include "client_amd/common.h"

#include <cstring>
#include <ostream>

namespace client_amd {

const Error Error::Success("");

std::ostream& operator<<(std::ostream& out, const Error& err) {
  if (!err.IsOk()) out << "error: " << err.Message();
  return out;
}

Error InferInput::Create(
    InferInput** infer_input, const std::string& name,
    const std::vector<int64_t>& dims, const std::string& datatype) {
  *infer_input = new InferInput(name, dims, datatype);
  return Error::Success;
}

Error InferInput::AppendFromString(const std::vector<std::string>& input) {
  // Serialize to the BYTES wire form (4-byte LE length + payload) and
  // keep ownership of the backing storage (reference common.h:306-318).
  str_bufs_.emplace_back();
  std::string& buf = str_bufs_.back();
  for (const auto& s : input) {
    uint32_t len = (uint32_t)s.size();
    buf.append(reinterpret_cast<const char*>(&len), 4);
    buf.append(s);
  }
  return AppendRaw(
      reinterpret_cast<const uint8_t*>(buf.data()), buf.size());
}

Error InferRequestedOutput::Create(
    InferRequestedOutput** infer_output, const std::string& name,
    const size_t class_count) {
  *infer_output = new InferRequestedOutput(name, class_count);
  return Error::Success;
}

Error InferResult::StringData(
    const std::string& output_name,
    std::vector<std::string>* string_result) const {
  const uint8_t* buf;
  size_t byte_size;
  RETURN_IF_ERROR(RawData(output_name, &buf, &byte_size));
  string_result->clear();
  size_t pos = 0;
  while (pos + 4 <= byte_size) {
    uint32_t len;
    memcpy(&len, buf + pos, 4);
    pos += 4;
    if (pos + len > byte_size) return Error("malformed BYTES tensor");
    string_result->emplace_back(
        reinterpret_cast<const char*>(buf + pos), len);
    pos += len;
  }
  return Error::Success;
}

Error InferenceServerClient::UpdateInferStat(const RequestTimers& timer) {
  // validate: all durations present and sane (reference common.cc:55-)
  const uint64_t request_ns = timer.Duration(
      RequestTimers::Kind::REQUEST_START, RequestTimers::Kind::REQUEST_END);
  if (request_ns == UINT64_MAX) {
    return Error("Timer not set correctly.");
  }
  uint64_t send_ns = timer.Duration(
      RequestTimers::Kind::SEND_START, RequestTimers::Kind::SEND_END);
  if (send_ns == UINT64_MAX) send_ns = 0;
  uint64_t recv_ns = timer.Duration(
      RequestTimers::Kind::RECV_START, RequestTimers::Kind::RECV_END);
  if (recv_ns == UINT64_MAX) recv_ns = 0;

  std::lock_guard<std::mutex> lock(stat_mu_);
  infer_stat_.completed_request_count++;
  infer_stat_.cumulative_total_request_time_ns += request_ns;
  infer_stat_.cumulative_send_time_ns += send_ns;
  infer_stat_.cumulative_receive_time_ns += recv_ns;
  return Error::Success;
}

}  // namespace client_amd
