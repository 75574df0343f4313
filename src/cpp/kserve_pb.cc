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

ModelTensorConfigPb DecodeTensorConfig(const uint8_t* data, size_t n,
                                       bool is_input) {
  ModelTensorConfigPb t;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: t.name = r.str(); break;
      case 2: t.data_type = (int32_t)r.varint(); break;
      case 3:
        // ModelInput.format (varint) vs ModelOutput.dims (packed)
        if (is_input) r.skip(wire);
        else DecodePackedOrSingleI64(&r, wire, &t.dims);
        break;
      case 4:
        if (is_input) DecodePackedOrSingleI64(&r, wire, &t.dims);
        else t.label_filename = r.str();
        break;
      case 6: t.is_shape_tensor = r.varint() != 0; break;
      case 8:
        if (is_input) t.optional_input = r.varint() != 0;
        else r.skip(wire);
        break;
      default: r.skip(wire);
    }
  }
  return t;
}

ModelInstanceGroupPb DecodeInstanceGroup(const uint8_t* data, size_t n) {
  ModelInstanceGroupPb g;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: g.name = r.str(); break;
      case 2: g.count = (int32_t)r.varint(); break;
      case 3: DecodePackedOrSingleI32(&r, wire, &g.gpus); break;
      case 4: g.kind = (int32_t)r.varint(); break;
      default: r.skip(wire);
    }
  }
  return g;
}

ModelDynamicBatchingPb DecodeDynBatch(const uint8_t* data, size_t n) {
  ModelDynamicBatchingPb d;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: DecodePackedOrSingleI32(&r, wire, &d.preferred_batch_size); break;
      case 2: d.max_queue_delay_microseconds = r.varint(); break;
      case 3: d.preserve_ordering = r.varint() != 0; break;
      default: r.skip(wire);
    }
  }
  return d;
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
  ModelConfigPb c;
  Reader r(data, n);
  int field, wire;
  while (r.next(&field, &wire)) {
    switch (field) {
      case 1: c.name = r.str(); break;
      case 2: c.platform = r.str(); break;
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
      case 11: {
        auto [ptr, len] = r.bytes();
        c.has_dynamic_batching = true;
        c.dynamic_batching = DecodeDynBatch(ptr, len);
        break;
      }
      case 13: c.has_sequence_batching = true; r.skip(wire); break;
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
          } else {
            sub.skip(w2);
          }
        }
        break;
      }
      case 17: c.backend = r.str(); break;
      case 19: {
        auto [ptr, len] = r.bytes();
        Reader sub(ptr, len);
        int f2, w2;
        while (sub.next(&f2, &w2)) {
          if (f2 == 1) c.decoupled = sub.varint() != 0;
          else sub.skip(w2);
        }
        break;
      }
      case 24: {
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
