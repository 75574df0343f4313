// Cross-decode oracle tool: reads a serialized ModelConfigResponse on
// stdin, decodes it with the C++ kserve_pb full-tree decoder, and dumps
// the result as JSON. tests/test_cpp_client.py builds a maximal config
// with the Python runtime schema (client_amd/grpc/_proto.py — the
// complete model_config.proto message tree) and compares field by
// field, proving the two stacks agree on the wire format.

#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "client_amd/kserve_pb.h"

using namespace client_amd::kserve;

static std::string esc(const std::string& s) {
  std::string out;
  for (char ch : s) {
    if (ch == '"' || ch == '\\') {
      out += '\\';
      out += ch;
    } else if ((unsigned char)ch < 0x20) {
      char buf[8];
      snprintf(buf, sizeof(buf), "\\u%04x", ch);
      out += buf;
    } else {
      out += ch;
    }
  }
  return out;
}

struct J {
  std::string s = "{";
  bool first = true;

  void comma() {
    if (!first) s += ",";
    first = false;
  }
  void kv(const std::string& k, const std::string& v) {
    comma();
    s += "\"" + esc(k) + "\":\"" + esc(v) + "\"";
  }
  void kn(const std::string& k, long long v) {
    comma();
    s += "\"" + esc(k) + "\":" + std::to_string(v);
  }
  void kd(const std::string& k, double v) {
    comma();
    char buf[64];
    snprintf(buf, sizeof(buf), "%.17g", v);
    s += "\"" + esc(k) + "\":" + buf;
  }
  void kb(const std::string& k, bool v) {
    comma();
    s += "\"" + esc(k) + "\":" + (v ? "true" : "false");
  }
  void raw(const std::string& k, const std::string& json) {
    comma();
    s += "\"" + esc(k) + "\":" + json;
  }
  std::string close() { return s + "}"; }
};

template <typename T, typename F>
static std::string arr(const std::vector<T>& v, F f) {
  std::string out = "[";
  for (size_t i = 0; i < v.size(); ++i) {
    if (i) out += ",";
    out += f(v[i]);
  }
  return out + "]";
}

static std::string num_arr_i64(const std::vector<int64_t>& v) {
  return arr(v, [](int64_t x) { return std::to_string(x); });
}
static std::string num_arr_i32(const std::vector<int32_t>& v) {
  return arr(v, [](int32_t x) { return std::to_string(x); });
}
static std::string str_arr(const std::vector<std::string>& v) {
  return arr(v, [](const std::string& x) {
    return std::string("\"") + esc(x) + "\"";
  });
}
static std::string str_map(const std::map<std::string, std::string>& m) {
  J j;
  for (const auto& [k, v] : m) j.kv(k, v);
  return j.close();
}

static std::string tensor_json(const ModelTensorConfigPb& t) {
  J j;
  j.kv("name", t.name);
  j.kn("data_type", t.data_type);
  j.raw("dims", num_arr_i64(t.dims));
  j.kb("has_reshape", t.has_reshape);
  j.raw("reshape", num_arr_i64(t.reshape.shape));
  j.kv("label_filename", t.label_filename);
  j.kb("is_shape_tensor", t.is_shape_tensor);
  j.kb("is_non_linear_format_io", t.is_non_linear_format_io);
  j.kn("format", t.format);
  j.kb("allow_ragged_batch", t.allow_ragged_batch);
  j.kb("optional", t.optional_input);
  return j.close();
}

static std::string group_json(const ModelInstanceGroupPb& g) {
  J j;
  j.kv("name", g.name);
  j.kn("kind", g.kind);
  j.kn("count", g.count);
  j.raw("gpus", num_arr_i32(g.gpus));
  j.raw("profile", str_arr(g.profile));
  j.kb("passive", g.passive);
  j.kv("host_policy", g.host_policy);
  j.kb("has_rate_limiter", g.has_rate_limiter);
  j.kn("rate_limiter_priority", g.rate_limiter.priority);
  j.raw("rate_limiter_resources",
        arr(g.rate_limiter.resources,
            [](const ModelRateLimiterPb::Resource& res) {
              J r;
              r.kv("name", res.name);
              r.kb("global", res.global);
              r.kn("count", res.count);
              return r.close();
            }));
  j.raw("secondary_devices",
        arr(g.secondary_devices,
            [](const ModelInstanceGroupPb::SecondaryDevice& d) {
              J r;
              r.kn("kind", d.kind);
              r.kn("device_id", d.device_id);
              return r.close();
            }));
  return j.close();
}

int main() {
  std::vector<uint8_t> buf;
  char tmp[4096];
  size_t got;
  while ((got = fread(tmp, 1, sizeof(tmp), stdin)) > 0) {
    buf.insert(buf.end(), tmp, tmp + got);
  }
  ModelConfigPb c = ModelConfigPb::Decode(buf.data(), buf.size());

  J j;
  j.kv("name", c.name);
  j.kv("platform", c.platform);
  j.kv("backend", c.backend);
  j.kv("runtime", c.runtime);
  j.kn("max_batch_size", c.max_batch_size);
  j.kn("version_policy_choice", (int)c.version_policy.choice);
  j.kn("version_policy_latest", c.version_policy.latest_num_versions);
  j.raw("version_policy_specific",
        num_arr_i64(c.version_policy.specific_versions));
  j.raw("input", arr(c.input, tensor_json));
  j.raw("output", arr(c.output, tensor_json));
  j.raw("instance_group", arr(c.instance_group, group_json));
  j.kv("default_model_filename", c.default_model_filename);
  j.raw("cc_model_filenames", str_map(c.cc_model_filenames));
  j.raw("metric_tags", str_map(c.metric_tags));
  j.raw("parameters", str_map(c.parameters));

  j.kb("has_optimization", c.has_optimization);
  {
    const auto& o = c.optimization;
    J oj;
    oj.kb("has_graph", o.has_graph);
    oj.kn("graph_level", o.graph_level);
    oj.kn("priority", o.priority);
    oj.kb("has_cuda", o.has_cuda);
    oj.kb("cuda_graphs", o.cuda_graphs);
    oj.kb("cuda_busy_wait_events", o.cuda_busy_wait_events);
    oj.kb("cuda_output_copy_stream", o.cuda_output_copy_stream);
    oj.raw("cuda_graph_spec",
           arr(o.cuda_graph_spec,
               [](const ModelOptimizationPolicyPb::GraphSpec& g) {
                 J gj;
                 gj.kn("batch_size", g.batch_size);
                 gj.kb("has_lower_bound", g.has_lower_bound);
                 gj.kn("lower_bound_batch_size", g.lower_bound_batch_size);
                 J im;
                 for (const auto& [k, v] : g.input)
                   im.raw(k, num_arr_i64(v.dim));
                 gj.raw("input", im.close());
                 return gj.close();
               }));
    auto accel = [](const std::vector<ModelOptimizationPolicyPb::Accelerator>&
                        v) {
      return arr(v, [](const ModelOptimizationPolicyPb::Accelerator& a) {
        J aj;
        aj.kv("name", a.name);
        aj.raw("parameters", str_map(a.parameters));
        return aj.close();
      });
    };
    oj.kb("has_execution_accelerators", o.has_execution_accelerators);
    oj.raw("gpu_execution_accelerator", accel(o.gpu_execution_accelerator));
    oj.raw("cpu_execution_accelerator", accel(o.cpu_execution_accelerator));
    oj.kb("has_input_pinned_memory", o.has_input_pinned_memory);
    oj.kb("input_pinned_memory", o.input_pinned_memory);
    oj.kb("has_output_pinned_memory", o.has_output_pinned_memory);
    oj.kb("output_pinned_memory", o.output_pinned_memory);
    oj.kn("gather_kernel_buffer_threshold",
          o.gather_kernel_buffer_threshold);
    oj.kb("eager_batching", o.eager_batching);
    j.raw("optimization", oj.close());
  }

  j.kb("has_dynamic_batching", c.has_dynamic_batching);
  {
    const auto& d = c.dynamic_batching;
    J dj;
    dj.raw("preferred_batch_size", num_arr_i32(d.preferred_batch_size));
    dj.kn("max_queue_delay_microseconds",
          (long long)d.max_queue_delay_microseconds);
    dj.kb("preserve_ordering", d.preserve_ordering);
    dj.kn("priority_levels", (long long)d.priority_levels);
    dj.kn("default_priority_level", (long long)d.default_priority_level);
    dj.kb("has_default_queue_policy", d.has_default_queue_policy);
    auto qp = [](const ModelQueuePolicyPb& q) {
      J qj;
      qj.kn("timeout_action", q.timeout_action);
      qj.kn("default_timeout_microseconds",
            (long long)q.default_timeout_microseconds);
      qj.kb("allow_timeout_override", q.allow_timeout_override);
      qj.kn("max_queue_size", q.max_queue_size);
      return qj.close();
    };
    dj.raw("default_queue_policy", qp(d.default_queue_policy));
    J pm;
    for (const auto& [k, v] : d.priority_queue_policy)
      pm.raw(std::to_string(k), qp(v));
    dj.raw("priority_queue_policy", pm.close());
    j.raw("dynamic_batching", dj.close());
  }

  j.kb("has_sequence_batching", c.has_sequence_batching);
  {
    const auto& s = c.sequence_batching;
    J sj;
    sj.kn("strategy", (int)s.strategy);
    sj.kn("direct_max_queue_delay_microseconds",
          (long long)s.direct_max_queue_delay_microseconds);
    sj.kd("direct_minimum_slot_utilization",
          s.direct_minimum_slot_utilization);
    sj.kn("oldest_max_candidate_sequences",
          s.oldest_max_candidate_sequences);
    sj.raw("oldest_preferred_batch_size",
           num_arr_i32(s.oldest_preferred_batch_size));
    sj.kn("oldest_max_queue_delay_microseconds",
          (long long)s.oldest_max_queue_delay_microseconds);
    sj.kb("oldest_preserve_ordering", s.oldest_preserve_ordering);
    sj.kn("max_sequence_idle_microseconds",
          (long long)s.max_sequence_idle_microseconds);
    sj.kb("iterative_sequence", s.iterative_sequence);
    sj.raw("control_input",
           arr(s.control_input,
               [](const ModelSequenceBatchingPb::ControlInput& ci) {
                 J cj;
                 cj.kv("name", ci.name);
                 cj.raw("control",
                        arr(ci.control,
                            [](const ModelSequenceBatchingPb::Control& c2) {
                              J kj;
                              kj.kn("kind", c2.kind);
                              kj.kn("data_type", c2.data_type);
                              kj.raw("int32_false_true",
                                     num_arr_i32(c2.int32_false_true));
                              kj.raw("fp32_false_true",
                                     arr(c2.fp32_false_true, [](float f) {
                                       char b[64];
                                       snprintf(b, sizeof(b), "%.9g", f);
                                       return std::string(b);
                                     }));
                              kj.raw("bool_false_true",
                                     [&] {
                                       std::string out = "[";
                                       for (size_t i = 0;
                                            i < c2.bool_false_true.size();
                                            ++i) {
                                         if (i) out += ",";
                                         out += c2.bool_false_true[i]
                                                    ? "true" : "false";
                                       }
                                       return out + "]";
                                     }());
                              return kj.close();
                            }));
                 return cj.close();
               }));
    sj.raw("state",
           arr(s.state, [](const ModelSequenceBatchingPb::State& st) {
             J stj;
             stj.kv("input_name", st.input_name);
             stj.kv("output_name", st.output_name);
             stj.kn("data_type", st.data_type);
             stj.raw("dims", num_arr_i64(st.dims));
             stj.kb("use_same_buffer_for_input_output",
                    st.use_same_buffer_for_input_output);
             stj.kb("use_growable_memory", st.use_growable_memory);
             stj.raw(
                 "initial_state",
                 arr(st.initial_state,
                     [](const ModelSequenceBatchingPb::InitialState& is) {
                       J ij;
                       ij.kv("name", is.name);
                       ij.kn("data_type", is.data_type);
                       ij.raw("dims", num_arr_i64(is.dims));
                       ij.kn("data_choice", (int)is.data_choice);
                       ij.kb("zero_data", is.zero_data);
                       ij.kv("data_file", is.data_file);
                       return ij.close();
                     }));
             return stj.close();
           }));
    j.raw("sequence_batching", sj.close());
  }

  j.kb("has_ensemble_scheduling", c.has_ensemble_scheduling);
  j.kn("ensemble_max_inflight_requests", c.ensemble_max_inflight_requests);
  j.raw("ensemble_steps", arr(c.ensemble_steps, [](const EnsembleStepPb& e) {
          J ej;
          ej.kv("model_name", e.model_name);
          ej.kn("model_version", (long long)e.model_version);
          ej.raw("input_map", str_map(e.input_map));
          ej.raw("output_map", str_map(e.output_map));
          ej.kv("model_namespace", e.model_namespace);
          return ej.close();
        }));

  j.raw("model_warmup", arr(c.model_warmup, [](const ModelWarmupPb& w) {
          J wj;
          wj.kv("name", w.name);
          wj.kn("batch_size", w.batch_size);
          wj.kn("count", w.count);
          J im;
          for (const auto& [k, v] : w.inputs) {
            J ij;
            ij.kn("data_type", v.data_type);
            ij.raw("dims", num_arr_i64(v.dims));
            ij.kn("data_choice", (int)v.data_choice);
            ij.kb("zero_data", v.zero_data);
            ij.kb("random_data", v.random_data);
            ij.kv("input_data_file", v.input_data_file);
            im.raw(k, ij.close());
          }
          wj.raw("inputs", im.close());
          return wj.close();
        }));

  j.raw("batch_input", arr(c.batch_input, [](const BatchInputPb& b) {
          J bj;
          bj.kn("kind", b.kind);
          bj.kn("data_type", b.data_type);
          bj.raw("target_name", str_arr(b.target_name));
          bj.raw("source_input", str_arr(b.source_input));
          return bj.close();
        }));
  j.raw("batch_output", arr(c.batch_output, [](const BatchOutputPb& b) {
          J bj;
          bj.kn("kind", b.kind);
          bj.raw("target_name", str_arr(b.target_name));
          bj.raw("source_input", str_arr(b.source_input));
          return bj.close();
        }));

  j.raw("op_library_filename", str_arr(c.op_library_filename));
  j.kb("decoupled", c.decoupled);
  j.raw("repository_agents",
        arr(c.repository_agents, [](const ModelRepositoryAgentPb& a) {
          J aj;
          aj.kv("name", a.name);
          aj.raw("parameters", str_map(a.parameters));
          return aj.close();
        }));
  j.kb("response_cache_enable", c.response_cache_enable);
  j.raw("metric_control",
        arr(c.metric_control, [](const ModelMetricControlPb& m) {
          J mj;
          mj.kv("family", m.family);
          mj.raw("histogram_buckets",
                 arr(m.histogram_buckets, [](double d) {
                   char b[64];
                   snprintf(b, sizeof(b), "%.17g", d);
                   return std::string(b);
                 }));
          return mj.close();
        }));

  printf("%s\n", j.close().c_str());
  return 0;
}
