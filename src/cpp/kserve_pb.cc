#include "client_amd/kserve_pb.h"

namespace client_amd {
namespace kserve {

using pb::Reader;
using pb::Writer;

std::string InferParameter::Encode() const {
  Writer w;
  // oneof members are serialized even at their default value
  switch (kind) {
    case BOOL: w.put_uint_always(1, b ? 1 : 0); break;
    case INT64: w.put_uint_always(2, (uint64_t)i); break;
    case STRING: w.tag(3, pb::LEN); w.varint(s.size()); w.out += s; break;
    case DOUBLE: {
      w.tag(4, pb::I64);
      uint64_t bits;
      memcpy(&bits, &d, 8);
      for (int k = 0; k < 8; ++k) w.out.push_back((char)((bits >> (8 * k)) & 0xFF));
      break;
    }
    case UINT64: w.put_uint_always(5, u); break;
    case NONE: break;
  }
  return w.out;
}

InferParameter InferParameter::Decode(const uint8_t* data, size_t n) {
  InferParameter p;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: p.kind = BOOL; p.b = r.varint() != 0; break;
      case 2: p.kind = INT64; p.i = (int64_t)r.varint(); break;
      case 3: p.kind = STRING; p.s = r.str(); break;
      case 4: {
        p.kind = DOUBLE;
        uint64_t bits = r.fixed64();
        memcpy(&p.d, &bits, 8);
        break;
      }
      case 5: p.kind = UINT64; p.u = r.varint(); break;
      default: r.skip(wire);
    }
  }
  return p;
}

std::string EncodeParamMapEntry(const std::string& key,
                                const InferParameter& value) {
  Writer w;
  w.put_str(1, key);
  w.put_msg(2, value.Encode());
  return w.out;
}

void DecodeParamMapEntry(const uint8_t* data, size_t n, ParamMap* out) {
  Reader r(data, n);
  int field, wire;
  std::string key;
  InferParameter value;
  while (r.next(&field, &wire)) {
    if (field == 1 && wire == pb::LEN) {
      key = r.str();
    } else if (field == 2 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      value = InferParameter::Decode(ptr, len);
    } else {
      r.skip(wire);
    }
  }
  (*out)[key] = value;
}

static void EncodeParams(Writer* w, int field, const ParamMap& params) {
  for (const auto& kv : params) {
    w->put_msg(field, EncodeParamMapEntry(kv.first, kv.second));
  }
}

std::string InferInputTensorPb::Encode() const {
  Writer w;
  w.put_str(1, name);
  w.put_str(2, datatype);
  w.put_packed_i64(3, shape);
  EncodeParams(&w, 4, parameters);
  return w.out;
}

std::string InferRequestedOutputPb::Encode() const {
  Writer w;
  w.put_str(1, name);
  EncodeParams(&w, 2, parameters);
  return w.out;
}

std::string ModelInferRequestPb::Encode() const {
  Writer w;
  w.put_str(1, model_name);
  w.put_str(2, model_version);
  w.put_str(3, id);
  EncodeParams(&w, 4, parameters);
  for (const auto& in : inputs) w.put_msg(5, in.Encode());
  for (const auto& out : outputs) w.put_msg(6, out.Encode());
  for (const auto& raw : raw_input_contents)
    w.put_bytes(7, raw.data(), raw.size());
  return w.out;
}

InferOutputTensorPb InferOutputTensorPb::Decode(const uint8_t* data,
                                                size_t n) {
  InferOutputTensorPb t;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: t.name = r.str(); break;
      case 2: t.datatype = r.str(); break;
      case 3:
        if (wire == pb::LEN) {
          t.shape = r.packed_i64();
        } else {
          t.shape.push_back((int64_t)r.varint());
        }
        break;
      case 4: {
        auto [ptr, len] = r.bytes();
        DecodeParamMapEntry(ptr, len, &t.parameters);
        break;
      }
      default: r.skip(wire);
    }
  }
  return t;
}

ModelInferResponsePb ModelInferResponsePb::Decode(const uint8_t* data,
                                                  size_t n) {
  ModelInferResponsePb m;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: m.model_name = r.str(); break;
      case 2: m.model_version = r.str(); break;
      case 3: m.id = r.str(); break;
      case 4: {
        auto [ptr, len] = r.bytes();
        DecodeParamMapEntry(ptr, len, &m.parameters);
        break;
      }
      case 5: {
        auto [ptr, len] = r.bytes();
        m.outputs.push_back(InferOutputTensorPb::Decode(ptr, len));
        break;
      }
      case 6: m.raw_output_contents.push_back(r.str()); break;
      default: r.skip(wire);
    }
  }
  return m;
}

ModelStreamInferResponsePb ModelStreamInferResponsePb::Decode(
    const uint8_t* data, size_t n) {
  ModelStreamInferResponsePb m;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1 && wire == pb::LEN) {
      m.error_message = r.str();
    } else if (field == 2 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      m.infer_response = ModelInferResponsePb::Decode(ptr, len);
    } else {
      r.skip(wire);
    }
  }
  return m;
}

std::string EncodeEmpty() { return std::string(); }

std::string EncodeNameVersion(const std::string& name,
                              const std::string& version) {
  Writer w;
  w.put_str(1, name);
  w.put_str(2, version);
  return w.out;
}

std::string EncodeName(const std::string& name) {
  Writer w;
  w.put_str(1, name);
  return w.out;
}

bool DecodeBoolField1(const uint8_t* data, size_t n) {
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1 && wire == pb::VARINT) return r.varint() != 0;
    r.skip(wire);
  }
  return false;
}

ServerMetadataPb ServerMetadataPb::Decode(const uint8_t* data, size_t n) {
  ServerMetadataPb m;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: m.name = r.str(); break;
      case 2: m.version = r.str(); break;
      case 3: m.extensions.push_back(r.str()); break;
      default: r.skip(wire);
    }
  }
  return m;
}

static TensorMetadataPb DecodeTensorMeta(const uint8_t* data, size_t n) {
  TensorMetadataPb t;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: t.name = r.str(); break;
      case 2: t.datatype = r.str(); break;
      case 3:
        if (wire == pb::LEN) t.shape = r.packed_i64();
        else t.shape.push_back((int64_t)r.varint());
        break;
      default: r.skip(wire);
    }
  }
  return t;
}

ModelMetadataPb ModelMetadataPb::Decode(const uint8_t* data, size_t n) {
  ModelMetadataPb m;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: m.name = r.str(); break;
      case 2: m.versions.push_back(r.str()); break;
      case 3: m.platform = r.str(); break;
      case 4: {
        auto [ptr, len] = r.bytes();
        m.inputs.push_back(DecodeTensorMeta(ptr, len));
        break;
      }
      case 5: {
        auto [ptr, len] = r.bytes();
        m.outputs.push_back(DecodeTensorMeta(ptr, len));
        break;
      }
      default: r.skip(wire);
    }
  }
  return m;
}

std::vector<RepositoryIndexEntryPb> DecodeRepositoryIndex(const uint8_t* data,
                                                          size_t n) {
  std::vector<RepositoryIndexEntryPb> out;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      Reader sub(ptr, len);
      RepositoryIndexEntryPb e;
      int f2, w2;
      while (sub.next(&f2, &w2)) {
        switch (f2) {
          case 1: e.name = sub.str(); break;
          case 2: e.version = sub.str(); break;
          case 3: e.state = sub.str(); break;
          case 4: e.reason = sub.str(); break;
          default: sub.skip(w2);
        }
      }
      out.push_back(std::move(e));
    } else {
      r.skip(wire);
    }
  }
  return out;
}

std::string EncodeRepositoryModelRequest(const std::string& model_name) {
  Writer w;
  w.put_str(2, model_name);
  return w.out;
}

std::string EncodeRepositoryModelLoadRequest(
    const std::string& model_name, const std::string& config,
    const std::map<std::string, std::string>& files) {
  // RepositoryModelLoadRequest { model_name = 2;
  //   map<string, ModelRepositoryParameter> parameters = 3; }
  // ModelRepositoryParameter oneof: string_param = 3, bytes_param = 4
  Writer w;
  w.put_str(2, model_name);
  auto put_param = [&w](const std::string& key, int oneof_field,
                        const std::string& val) {
    Writer param;
    param.put_str(oneof_field, val);
    Writer entry;
    entry.put_str(1, key);
    entry.put_msg(2, param.out);
    w.put_msg(3, entry.out);
  };
  if (!config.empty()) put_param("config", 3, config);
  for (const auto& [path, content] : files) put_param(path, 4, content);
  return w.out;
}

std::string EncodeSystemShmRegister(const std::string& name,
                                    const std::string& key, uint64_t offset,
                                    uint64_t byte_size) {
  Writer w;
  w.put_str(1, name);
  w.put_str(2, key);
  w.put_uint(3, offset);
  w.put_uint(4, byte_size);
  return w.out;
}

std::string EncodeCudaShmRegister(const std::string& name,
                                  const std::string& raw_handle,
                                  int64_t device_id, uint64_t byte_size) {
  Writer w;
  w.put_str(1, name);
  w.put_bytes(2, raw_handle.data(), raw_handle.size());
  w.put_uint(3, (uint64_t)device_id);
  w.put_uint(4, byte_size);
  return w.out;
}

static StatisticDurationPb DecodeDuration(const uint8_t* data, size_t n) {
  StatisticDurationPb d;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1) d.count = r.varint();
    else if (field == 2) d.ns = r.varint();
    else r.skip(wire);
  }
  return d;
}

std::vector<ModelStatisticsPb> DecodeModelStatistics(const uint8_t* data,
                                                     size_t n) {
  std::vector<ModelStatisticsPb> out;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      Reader sub(ptr, len);
      ModelStatisticsPb m;
      int f2, w2;
      while (sub.next(&f2, &w2)) {
        switch (f2) {
          case 1: m.name = sub.str(); break;
          case 2: m.version = sub.str(); break;
          case 3: m.last_inference = sub.varint(); break;
          case 4: m.inference_count = sub.varint(); break;
          case 5: m.execution_count = sub.varint(); break;
          case 6: {
            auto [p2, l2] = sub.bytes();
            Reader st(p2, l2);
            int f3, w3;
            while (st.next(&f3, &w3)) {
              auto [p3, l3] = st.bytes();
              StatisticDurationPb d = DecodeDuration(p3, l3);
              switch (f3) {
                case 1: m.success = d; break;
                case 2: m.fail = d; break;
                case 3: m.queue = d; break;
                case 4: m.compute_input = d; break;
                case 5: m.compute_infer = d; break;
                case 6: m.compute_output = d; break;
                default: break;
              }
            }
            break;
          }
          default: sub.skip(w2);
        }
      }
      out.push_back(std::move(m));
    } else {
      r.skip(wire);
    }
  }
  return out;
}

//==============================================================================
// Trace settings

std::string EncodeTraceSettingRequest(const TraceSettingsPb& settings,
                                      const std::string& model_name) {
  pb::Writer w;
  for (const auto& kv : settings) {
    pb::Writer val;  // SettingValue { repeated string value = 1; }
    for (const auto& v : kv.second) val.put_str(1, v);
    pb::Writer entry;
    entry.put_str(1, kv.first);
    entry.put_msg(2, val.out);
    w.put_msg(1, entry.out);
  }
  w.put_str(2, model_name);
  return w.out;
}

TraceSettingsPb DecodeTraceSettingResponse(const uint8_t* data, size_t n) {
  TraceSettingsPb out;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      Reader entry(ptr, len);
      std::string key;
      std::vector<std::string> values;
      int f2, w2;
      while (entry.next(&f2, &w2)) {
        if (f2 == 1) {
          key = entry.str();
        } else if (f2 == 2 && w2 == pb::LEN) {
          auto [vp, vl] = entry.bytes();
          Reader val(vp, vl);
          int f3, w3;
          while (val.next(&f3, &w3)) {
            if (f3 == 1) values.push_back(val.str());
            else val.skip(w3);
          }
        } else {
          entry.skip(w2);
        }
      }
      out[key] = std::move(values);
    } else {
      r.skip(wire);
    }
  }
  return out;
}

//==============================================================================
// ModelConfig decode (subset; see kserve_pb.h).

namespace {

void DecodePackedOrSingleI32(Reader* r, int wire, std::vector<int32_t>* out) {
  if (wire == pb::LEN) {
    auto [ptr, len] = r->bytes();
    Reader sub(ptr, len);
    while (!sub.done()) out->push_back((int32_t)sub.varint());
  } else {
    out->push_back((int32_t)r->varint());
  }
}

void DecodePackedOrSingleI64(Reader* r, int wire, std::vector<int64_t>* out) {
  if (wire == pb::LEN) {
    auto [ptr, len] = r->bytes();
    Reader sub(ptr, len);
    while (!sub.done()) out->push_back((int64_t)sub.varint());
  } else {
    out->push_back((int64_t)r->varint());
  }
}

void DecodePackedOrSingleFloat(Reader* r, int wire, std::vector<float>* out) {
  auto as_float = [](uint32_t u) {
    float f;
    std::memcpy(&f, &u, 4);
    return f;
  };
  if (wire == pb::LEN) {
    auto [ptr, len] = r->bytes();
    Reader sub(ptr, len);
    while (!sub.done()) out->push_back(as_float(sub.fixed32()));
  } else {
    out->push_back(as_float(r->fixed32()));
  }
}

void DecodePackedOrSingleDouble(Reader* r, int wire,
                                std::vector<double>* out) {
  auto as_double = [](uint64_t u) {
    double d;
    std::memcpy(&d, &u, 8);
    return d;
  };
  if (wire == pb::LEN) {
    auto [ptr, len] = r->bytes();
    Reader sub(ptr, len);
    while (!sub.done()) out->push_back(as_double(sub.fixed64()));
  } else {
    out->push_back(as_double(r->fixed64()));
  }
}

void DecodePackedOrSingleBool(Reader* r, int wire, std::vector<bool>* out) {
  if (wire == pb::LEN) {
    auto [ptr, len] = r->bytes();
    Reader sub(ptr, len);
    while (!sub.done()) out->push_back(sub.varint() != 0);
  } else {
    out->push_back(r->varint() != 0);
  }
}

ModelTensorReshapePb DecodeReshape(const uint8_t* data, size_t n) {
  ModelTensorReshapePb re;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1) DecodePackedOrSingleI64(&r, wire, &re.shape);
    else r.skip(wire);
  }
  return re;
}

ModelTensorConfigPb DecodeTensorConfig(const uint8_t* data, size_t n,
                                       bool is_input) {
  // ModelInput (model_config.proto:317) / ModelOutput (:428)
  ModelTensorConfigPb t;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: t.name = r.str(); break;
      case 2: t.data_type = (int32_t)r.varint(); break;
      case 3:
        // ModelInput.format (varint) vs ModelOutput.dims (packed)
        if (is_input) t.format = (int32_t)r.varint();
        else DecodePackedOrSingleI64(&r, wire, &t.dims);
        break;
      case 4:
        if (is_input) DecodePackedOrSingleI64(&r, wire, &t.dims);
        else t.label_filename = r.str();
        break;
      case 5: {
        auto [ptr, len] = r.bytes();
        t.reshape = DecodeReshape(ptr, len);
        t.has_reshape = true;
        break;
      }
      case 6: t.is_shape_tensor = r.varint() != 0; break;
      case 7:
        if (is_input) t.allow_ragged_batch = r.varint() != 0;
        else t.is_non_linear_format_io = r.varint() != 0;
        break;
      case 8:
        if (is_input) t.optional_input = r.varint() != 0;
        else r.skip(wire);
        break;
      case 9:
        if (is_input) t.is_non_linear_format_io = r.varint() != 0;
        else r.skip(wire);
        break;
      default: r.skip(wire);
    }
  }
  return t;
}

ModelRateLimiterPb DecodeRateLimiter(const uint8_t* data, size_t n) {
  ModelRateLimiterPb rl;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: {
        auto [ptr, len] = r.bytes();
        ModelRateLimiterPb::Resource res;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          switch (f2) {
            case 1: res.name = sub.str(); break;
            case 2: res.global = sub.varint() != 0; break;
            case 3: res.count = (uint32_t)sub.varint(); break;
            default: sub.skip(w2);
          }
        }
        rl.resources.push_back(std::move(res));
        break;
      }
      case 2: rl.priority = (uint32_t)r.varint(); break;
      default: r.skip(wire);
    }
  }
  return rl;
}

ModelInstanceGroupPb DecodeInstanceGroup(const uint8_t* data, size_t n) {
  // model_config.proto:143
  ModelInstanceGroupPb g;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: g.name = r.str(); break;
      case 2: g.count = (int32_t)r.varint(); break;
      case 3: DecodePackedOrSingleI32(&r, wire, &g.gpus); break;
      case 4: g.kind = (int32_t)r.varint(); break;
      case 5: g.profile.push_back(r.str()); break;
      case 6: {
        auto [ptr, len] = r.bytes();
        g.rate_limiter = DecodeRateLimiter(ptr, len);
        g.has_rate_limiter = true;
        break;
      }
      case 7: g.passive = r.varint() != 0; break;
      case 8: {
        auto [ptr, len] = r.bytes();
        ModelInstanceGroupPb::SecondaryDevice sd;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) sd.kind = (int32_t)sub.varint();
          else if (f2 == 2) sd.device_id = (int64_t)sub.varint();
          else sub.skip(w2);
        }
        g.secondary_devices.push_back(sd);
        break;
      }
      case 9: g.host_policy = r.str(); break;
      default: r.skip(wire);
    }
  }
  return g;
}

ModelQueuePolicyPb DecodeQueuePolicy(const uint8_t* data, size_t n) {
  ModelQueuePolicyPb q;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: q.timeout_action = (int32_t)r.varint(); break;
      case 2: q.default_timeout_microseconds = r.varint(); break;
      case 3: q.allow_timeout_override = r.varint() != 0; break;
      case 4: q.max_queue_size = (uint32_t)r.varint(); break;
      default: r.skip(wire);
    }
  }
  return q;
}

ModelDynamicBatchingPb DecodeDynBatch(const uint8_t* data, size_t n) {
  // model_config.proto:1122
  ModelDynamicBatchingPb d;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: DecodePackedOrSingleI32(&r, wire, &d.preferred_batch_size); break;
      case 2: d.max_queue_delay_microseconds = r.varint(); break;
      case 3: d.preserve_ordering = r.varint() != 0; break;
      case 4: d.priority_levels = r.varint(); break;
      case 5: d.default_priority_level = r.varint(); break;
      case 6: {
        auto [ptr, len] = r.bytes();
        d.default_queue_policy = DecodeQueuePolicy(ptr, len);
        d.has_default_queue_policy = true;
        break;
      }
      case 7: {
        // map<uint64, ModelQueuePolicy> entry
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        uint64_t key = 0;
        ModelQueuePolicyPb val;
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) key = sub.varint();
          else if (f2 == 2 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            val = DecodeQueuePolicy(p2, l2);
          } else sub.skip(w2);
        }
        d.priority_queue_policy[key] = val;
        break;
      }
      default: r.skip(wire);
    }
  }
  return d;
}

ModelVersionPolicyPb DecodeVersionPolicy(const uint8_t* data, size_t n) {
  // oneof policy_choice (model_config.proto:635)
  ModelVersionPolicyPb vp;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: {  // Latest
        auto [ptr, len] = r.bytes();
        vp.choice = ModelVersionPolicyPb::LATEST;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) vp.latest_num_versions = (uint32_t)sub.varint();
          else sub.skip(w2);
        }
        break;
      }
      case 2:  // All
        vp.choice = ModelVersionPolicyPb::ALL;
        r.skip(wire);
        break;
      case 3: {  // Specific
        auto [ptr, len] = r.bytes();
        vp.choice = ModelVersionPolicyPb::SPECIFIC;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) DecodePackedOrSingleI64(&sub, w2,
                                              &vp.specific_versions);
          else sub.skip(w2);
        }
        break;
      }
      default: r.skip(wire);
    }
  }
  return vp;
}

void DecodeAccelerator(const uint8_t* data, size_t n,
                       std::vector<ModelOptimizationPolicyPb::Accelerator>*
                           out);

void DecodeGraphShapeMapEntry(
    const uint8_t* data, size_t n,
    std::map<std::string, ModelOptimizationPolicyPb::GraphSpecShape>* out) {
  Reader r(data, n);
  std::string key;
  ModelOptimizationPolicyPb::GraphSpecShape shape;
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1) key = r.str();
    else if (field == 2 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      Reader sub(ptr, len);
      int f2, w2;
      while (sub.next(&f2, &w2)) {
        if (f2 == 1) DecodePackedOrSingleI64(&sub, w2, &shape.dim);
        else sub.skip(w2);
      }
    } else r.skip(wire);
  }
  (*out)[key] = shape;
}

ModelOptimizationPolicyPb DecodeOptimization(const uint8_t* data, size_t n) {
  // model_config.proto:707
  ModelOptimizationPolicyPb o;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: {  // Graph
        auto [ptr, len] = r.bytes();
        o.has_graph = true;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) o.graph_level = (int32_t)sub.varint();
          else sub.skip(w2);
        }
        break;
      }
      case 2: o.priority = (int32_t)r.varint(); break;
      case 3: {  // Cuda
        auto [ptr, len] = r.bytes();
        o.has_cuda = true;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          switch (f2) {
            case 1: o.cuda_graphs = sub.varint() != 0; break;
            case 2: o.cuda_busy_wait_events = sub.varint() != 0; break;
            case 3: {  // GraphSpec
              auto [p2, l2] = sub.bytes();
              ModelOptimizationPolicyPb::GraphSpec spec;
              Reader gs(p2, l2);
              int f3, w3;
              while (gs.next(&f3, &w3)) {
                switch (f3) {
                  case 1: spec.batch_size = (int32_t)gs.varint(); break;
                  case 2: {
                    auto [p3, l3] = gs.bytes();
                    DecodeGraphShapeMapEntry(p3, l3, &spec.input);
                    break;
                  }
                  case 3: {  // LowerBound
                    auto [p3, l3] = gs.bytes();
                    spec.has_lower_bound = true;
                    Reader lb(p3, l3);
                    int f4, w4;
                    while (lb.next(&f4, &w4)) {
                      if (f4 == 1)
                        spec.lower_bound_batch_size = (int32_t)lb.varint();
                      else if (f4 == 2 && w4 == pb::LEN) {
                        auto [p4, l4] = lb.bytes();
                        DecodeGraphShapeMapEntry(p4, l4,
                                                 &spec.lower_bound_input);
                      } else lb.skip(w4);
                    }
                    break;
                  }
                  default: gs.skip(w3);
                }
              }
              o.cuda_graph_spec.push_back(std::move(spec));
              break;
            }
            case 4: o.cuda_output_copy_stream = sub.varint() != 0; break;
            default: sub.skip(w2);
          }
        }
        break;
      }
      case 4: {  // ExecutionAccelerators
        auto [ptr, len] = r.bytes();
        o.has_execution_accelerators = true;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            DecodeAccelerator(p2, l2, &o.gpu_execution_accelerator);
          } else if (f2 == 2 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            DecodeAccelerator(p2, l2, &o.cpu_execution_accelerator);
          } else sub.skip(w2);
        }
        break;
      }
      case 5:
      case 6: {  // PinnedMemoryBuffer
        auto [ptr, len] = r.bytes();
        bool enable = false;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) enable = sub.varint() != 0;
          else sub.skip(w2);
        }
        if (field == 5) {
          o.input_pinned_memory = enable;
          o.has_input_pinned_memory = true;
        } else {
          o.output_pinned_memory = enable;
          o.has_output_pinned_memory = true;
        }
        break;
      }
      case 7: o.gather_kernel_buffer_threshold = (uint32_t)r.varint(); break;
      case 8: o.eager_batching = r.varint() != 0; break;
      default: r.skip(wire);
    }
  }
  return o;
}

void DecodeStrMapEntry(const uint8_t* data, size_t n,
                       std::map<std::string, std::string>* out);

void DecodeAccelerator(const uint8_t* data, size_t n,
                       std::vector<ModelOptimizationPolicyPb::Accelerator>*
                           out) {
  ModelOptimizationPolicyPb::Accelerator a;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1) a.name = r.str();
    else if (field == 2 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      DecodeStrMapEntry(ptr, len, &a.parameters);
    } else r.skip(wire);
  }
  out->push_back(std::move(a));
}

ModelSequenceBatchingPb DecodeSequenceBatching(const uint8_t* data,
                                               size_t n) {
  // model_config.proto:1197 — full strategy/control/state tree
  using SB = ModelSequenceBatchingPb;
  SB s;
  Reader r(data, n);
  int field, wire;
  auto decode_control = [](const uint8_t* p, size_t l) {
    SB::Control c;
    Reader sub(p, l);
    int f2, w2;
    while (sub.next(&f2, &w2)) {
      switch (f2) {
        case 1: c.kind = (int32_t)sub.varint(); break;
        case 2: DecodePackedOrSingleI32(&sub, w2, &c.int32_false_true); break;
        case 3: DecodePackedOrSingleFloat(&sub, w2, &c.fp32_false_true); break;
        case 4: c.data_type = (int32_t)sub.varint(); break;
        case 5: DecodePackedOrSingleBool(&sub, w2, &c.bool_false_true); break;
        default: sub.skip(w2);
      }
    }
    return c;
  };
  auto decode_initial_state = [](const uint8_t* p, size_t l) {
    SB::InitialState is;
    Reader sub(p, l);
    int f2, w2;
    while (sub.next(&f2, &w2)) {
      switch (f2) {
        case 1: is.data_type = (int32_t)sub.varint(); break;
        case 2: DecodePackedOrSingleI64(&sub, w2, &is.dims); break;
        case 3:
          is.data_choice = SB::InitialState::ZERO;
          is.zero_data = sub.varint() != 0;
          break;
        case 4:
          is.data_choice = SB::InitialState::FILE;
          is.data_file = sub.str();
          break;
        case 5: is.name = sub.str(); break;
        default: sub.skip(w2);
      }
    }
    return is;
  };
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: s.max_sequence_idle_microseconds = r.varint(); break;
      case 2: {  // ControlInput
        auto [ptr, len] = r.bytes();
        SB::ControlInput ci;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) ci.name = sub.str();
          else if (f2 == 2 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            ci.control.push_back(decode_control(p2, l2));
          } else sub.skip(w2);
        }
        s.control_input.push_back(std::move(ci));
        break;
      }
      case 3: {  // StrategyDirect
        auto [ptr, len] = r.bytes();
        s.strategy = SB::DIRECT;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) s.direct_max_queue_delay_microseconds = sub.varint();
          else if (f2 == 2 && w2 == pb::I32) {
            uint32_t u = sub.fixed32();
            std::memcpy(&s.direct_minimum_slot_utilization, &u, 4);
          } else sub.skip(w2);
        }
        break;
      }
      case 4: {  // StrategyOldest
        auto [ptr, len] = r.bytes();
        s.strategy = SB::OLDEST;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          switch (f2) {
            case 1:
              s.oldest_max_candidate_sequences = (int32_t)sub.varint();
              break;
            case 2:
              DecodePackedOrSingleI32(&sub, w2,
                                      &s.oldest_preferred_batch_size);
              break;
            case 3: s.oldest_max_queue_delay_microseconds = sub.varint();
              break;
            case 4: s.oldest_preserve_ordering = sub.varint() != 0; break;
            default: sub.skip(w2);
          }
        }
        break;
      }
      case 5: {  // State
        auto [ptr, len] = r.bytes();
        SB::State st;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          switch (f2) {
            case 1: st.input_name = sub.str(); break;
            case 2: st.output_name = sub.str(); break;
            case 3: st.data_type = (int32_t)sub.varint(); break;
            case 4: DecodePackedOrSingleI64(&sub, w2, &st.dims); break;
            case 5: {
              auto [p2, l2] = sub.bytes();
              st.initial_state.push_back(decode_initial_state(p2, l2));
              break;
            }
            case 6:
              st.use_same_buffer_for_input_output = sub.varint() != 0;
              break;
            case 7: st.use_growable_memory = sub.varint() != 0; break;
            default: sub.skip(w2);
          }
        }
        s.state.push_back(std::move(st));
        break;
      }
      case 6: s.iterative_sequence = r.varint() != 0; break;
      default: r.skip(wire);
    }
  }
  return s;
}

ModelWarmupPb DecodeWarmup(const uint8_t* data, size_t n) {
  // model_config.proto:1698
  ModelWarmupPb w;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: w.name = r.str(); break;
      case 2: w.batch_size = (uint32_t)r.varint(); break;
      case 3: {  // map<string, Input>
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        std::string key;
        ModelWarmupPb::Input in;
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) key = sub.str();
          else if (f2 == 2 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            Reader iv(p2, l2);
            int f3, w3;
            while (iv.next(&f3, &w3)) {
              switch (f3) {
                case 1: in.data_type = (int32_t)iv.varint(); break;
                case 2: DecodePackedOrSingleI64(&iv, w3, &in.dims); break;
                case 3:
                  in.data_choice = ModelWarmupPb::Input::ZERO;
                  in.zero_data = iv.varint() != 0;
                  break;
                case 4:
                  in.data_choice = ModelWarmupPb::Input::RANDOM;
                  in.random_data = iv.varint() != 0;
                  break;
                case 5:
                  in.data_choice = ModelWarmupPb::Input::FILE;
                  in.input_data_file = iv.str();
                  break;
                default: iv.skip(w3);
              }
            }
          } else sub.skip(w2);
        }
        w.inputs[key] = in;
        break;
      }
      case 4: w.count = (uint32_t)r.varint(); break;
      default: r.skip(wire);
    }
  }
  return w;
}

BatchInputPb DecodeBatchInput(const uint8_t* data, size_t n) {
  BatchInputPb b;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: b.kind = (int32_t)r.varint(); break;
      case 2: b.target_name.push_back(r.str()); break;
      case 3: b.data_type = (int32_t)r.varint(); break;
      case 4: b.source_input.push_back(r.str()); break;
      default: r.skip(wire);
    }
  }
  return b;
}

BatchOutputPb DecodeBatchOutput(const uint8_t* data, size_t n) {
  BatchOutputPb b;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: b.target_name.push_back(r.str()); break;
      case 2: b.kind = (int32_t)r.varint(); break;
      case 3: b.source_input.push_back(r.str()); break;
      default: r.skip(wire);
    }
  }
  return b;
}

void DecodeStrMapEntry(const uint8_t* data, size_t n,
                       std::map<std::string, std::string>* out) {
  Reader r(data, n);
  int field, wire;
  std::string k, v;
  while (r.next(&field, &wire)) {
    if (field == 1) k = r.str();
    else if (field == 2) v = r.str();
    else r.skip(wire);
  }
  (*out)[k] = v;
}

EnsembleStepPb DecodeEnsembleStep(const uint8_t* data, size_t n) {
  EnsembleStepPb e;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: e.model_name = r.str(); break;
      case 2: e.model_version = (int64_t)r.varint(); break;
      case 3: {
        auto [ptr, len] = r.bytes();
        DecodeStrMapEntry(ptr, len, &e.input_map);
        break;
      }
      case 4: {
        auto [ptr, len] = r.bytes();
        DecodeStrMapEntry(ptr, len, &e.output_map);
        break;
      }
      case 5: e.model_namespace = r.str(); break;
      default: r.skip(wire);
    }
  }
  return e;
}

void DecodeModelParamEntry(const uint8_t* data, size_t n,
                           std::map<std::string, std::string>* out) {
  Reader r(data, n);
  int field, wire;
  std::string k, v;
  while (r.next(&field, &wire)) {
    if (field == 1) {
      k = r.str();
    } else if (field == 2 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      Reader sub(ptr, len);
      int f2, w2;
      while (sub.next(&f2, &w2)) {
        if (f2 == 1) v = sub.str();
        else sub.skip(w2);
      }
    } else {
      r.skip(wire);
    }
  }
  (*out)[k] = v;
}

ModelConfigPb DecodeModelConfigMsg(const uint8_t* data, size_t n) {
  // the full ModelConfig message (model_config.proto:1971-2180)
  ModelConfigPb c;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: c.name = r.str(); break;
      case 2: c.platform = r.str(); break;
      case 3: {
        auto [ptr, len] = r.bytes();
        c.version_policy = DecodeVersionPolicy(ptr, len);
        break;
      }
      case 4: c.max_batch_size = (int32_t)r.varint(); break;
      case 5: {
        auto [ptr, len] = r.bytes();
        c.input.push_back(DecodeTensorConfig(ptr, len, true));
        break;
      }
      case 6: {
        auto [ptr, len] = r.bytes();
        c.output.push_back(DecodeTensorConfig(ptr, len, false));
        break;
      }
      case 7: {
        auto [ptr, len] = r.bytes();
        c.instance_group.push_back(DecodeInstanceGroup(ptr, len));
        break;
      }
      case 8: c.default_model_filename = r.str(); break;
      case 9: {
        auto [ptr, len] = r.bytes();
        DecodeStrMapEntry(ptr, len, &c.cc_model_filenames);
        break;
      }
      case 10: {
        auto [ptr, len] = r.bytes();
        DecodeStrMapEntry(ptr, len, &c.metric_tags);
        break;
      }
      case 11: {
        auto [ptr, len] = r.bytes();
        c.has_dynamic_batching = true;
        c.dynamic_batching = DecodeDynBatch(ptr, len);
        break;
      }
      case 12: {
        auto [ptr, len] = r.bytes();
        c.has_optimization = true;
        c.optimization = DecodeOptimization(ptr, len);
        break;
      }
      case 13: {
        auto [ptr, len] = r.bytes();
        c.has_sequence_batching = true;
        c.sequence_batching = DecodeSequenceBatching(ptr, len);
        break;
      }
      case 14: {
        auto [ptr, len] = r.bytes();
        DecodeModelParamEntry(ptr, len, &c.parameters);
        break;
      }
      case 15: {
        auto [ptr, len] = r.bytes();
        c.has_ensemble_scheduling = true;
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            c.ensemble_steps.push_back(DecodeEnsembleStep(p2, l2));
          } else if (f2 == 2) {
            c.ensemble_max_inflight_requests = (uint32_t)sub.varint();
          } else {
            sub.skip(w2);
          }
        }
        break;
      }
      case 16: {
        auto [ptr, len] = r.bytes();
        c.model_warmup.push_back(DecodeWarmup(ptr, len));
        break;
      }
      case 17: c.backend = r.str(); break;
      case 18: {  // ModelOperations
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) c.op_library_filename.push_back(sub.str());
          else sub.skip(w2);
        }
        break;
      }
      case 19: {  // ModelTransactionPolicy
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) c.decoupled = sub.varint() != 0;
          else sub.skip(w2);
        }
        break;
      }
      case 20: {
        auto [ptr, len] = r.bytes();
        c.batch_input.push_back(DecodeBatchInput(ptr, len));
        break;
      }
      case 21: {
        auto [ptr, len] = r.bytes();
        c.batch_output.push_back(DecodeBatchOutput(ptr, len));
        break;
      }
      case 23: {  // ModelRepositoryAgents
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            ModelRepositoryAgentPb agent;
            Reader av(p2, l2);
            int f3, w3;
            while (av.next(&f3, &w3)) {
              if (f3 == 1) agent.name = av.str();
              else if (f3 == 2 && w3 == pb::LEN) {
                auto [p3, l3] = av.bytes();
                DecodeStrMapEntry(p3, l3, &agent.parameters);
              } else av.skip(w3);
            }
            c.repository_agents.push_back(std::move(agent));
          } else sub.skip(w2);
        }
        break;
      }
      case 24: {  // ModelResponseCache
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) c.response_cache_enable = sub.varint() != 0;
          else sub.skip(w2);
        }
        break;
      }
      case 25: c.runtime = r.str(); break;
      case 26: {  // ModelMetrics
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1 && w2 == pb::LEN) {
            auto [p2, l2] = sub.bytes();
            ModelMetricControlPb mc;
            Reader mv(p2, l2);
            int f3, w3;
            while (mv.next(&f3, &w3)) {
              if (f3 == 1 && w3 == pb::LEN) {  // MetricIdentifier
                auto [p3, l3] = mv.bytes();
                Reader iv(p3, l3);
                int f4, w4;
                while (iv.next(&f4, &w4)) {
                  if (f4 == 1) mc.family = iv.str();
                  else iv.skip(w4);
                }
              } else if (f3 == 2 && w3 == pb::LEN) {  // HistogramOptions
                auto [p3, l3] = mv.bytes();
                Reader hv(p3, l3);
                int f4, w4;
                while (hv.next(&f4, &w4)) {
                  if (f4 == 1)
                    DecodePackedOrSingleDouble(&hv, w4,
                                               &mc.histogram_buckets);
                  else hv.skip(w4);
                }
              } else mv.skip(w3);
            }
            c.metric_control.push_back(std::move(mc));
          } else sub.skip(w2);
        }
        break;
      }
      default: r.skip(wire);
    }
  }
  return c;
}

}  // namespace

ModelConfigPb ModelConfigPb::Decode(const uint8_t* data, size_t n) {
  // ModelConfigResponse { ModelConfig config = 1; }
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    if (field == 1 && wire == pb::LEN) {
      auto [ptr, len] = r.bytes();
      return DecodeModelConfigMsg(ptr, len);
    }
    r.skip(wire);
  }
  return ModelConfigPb();
}

}  // namespace kserve
}  // namespace client_amd
