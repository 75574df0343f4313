// Hand-encoded KServe-v2 gRPC messages (field numbers per the vendored
// schema, grpc_service.proto — e.g. ModelInferRequest :575-706 with
// raw_input_contents = 7). Only the fields the client reads/writes are
// modeled; unknown fields are skipped on decode (forward compatible).
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

#include "client_amd/pb.h"

namespace client_amd {
namespace kserve {

// InferParameter oneof (bool=1, int64=2, string=3, double=4, uint64=5)
struct InferParameter {
  enum Kind { NONE, BOOL, INT64, STRING, DOUBLE, UINT64 } kind = NONE;
  bool b = false;
  int64_t i = 0;
  std::string s;
  double d = 0;
  uint64_t u = 0;

  static InferParameter Bool(bool v) {
    InferParameter p; p.kind = BOOL; p.b = v; return p;
  }
  static InferParameter Int(int64_t v) {
    InferParameter p; p.kind = INT64; p.i = v; return p;
  }
  static InferParameter Str(std::string v) {
    InferParameter p; p.kind = STRING; p.s = std::move(v); return p;
  }
  static InferParameter Uint(uint64_t v) {
    InferParameter p; p.kind = UINT64; p.u = v; return p;
  }

  std::string Encode() const;
  static InferParameter Decode(const uint8_t* data, size_t n);
};

using ParamMap = std::map<std::string, InferParameter>;

std::string EncodeParamMapEntry(const std::string& key,
                                const InferParameter& value);
void DecodeParamMapEntry(const uint8_t* data, size_t n, ParamMap* out);

struct InferInputTensorPb {
  std::string name;
  std::string datatype;
  std::vector<int64_t> shape;
  ParamMap parameters;
  std::string Encode() const;
};

struct InferRequestedOutputPb {
  std::string name;
  ParamMap parameters;
  std::string Encode() const;
};

struct ModelInferRequestPb {
  std::string model_name;
  std::string model_version;
  std::string id;
  ParamMap parameters;
  std::vector<InferInputTensorPb> inputs;
  std::vector<InferRequestedOutputPb> outputs;
  std::vector<std::string> raw_input_contents;
  std::string Encode() const;
};

struct InferOutputTensorPb {
  std::string name;
  std::string datatype;
  std::vector<int64_t> shape;
  ParamMap parameters;
  static InferOutputTensorPb Decode(const uint8_t* data, size_t n);
};

struct ModelInferResponsePb {
  std::string model_name;
  std::string model_version;
  std::string id;
  ParamMap parameters;
  std::vector<InferOutputTensorPb> outputs;
  std::vector<std::string> raw_output_contents;
  static ModelInferResponsePb Decode(const uint8_t* data, size_t n);
};

struct ModelStreamInferResponsePb {
  std::string error_message;
  ModelInferResponsePb infer_response;
  static ModelStreamInferResponsePb Decode(const uint8_t* data, size_t n);
};

// Simple request encoders / response decoders for the management RPCs.
std::string EncodeEmpty();
std::string EncodeNameVersion(const std::string& name,
                              const std::string& version);
std::string EncodeName(const std::string& name);  // field 1 string
bool DecodeBoolField1(const uint8_t* data, size_t n);  // live/ready

struct ServerMetadataPb {
  std::string name;
  std::string version;
  std::vector<std::string> extensions;
  static ServerMetadataPb Decode(const uint8_t* data, size_t n);
};

struct TensorMetadataPb {
  std::string name;
  std::string datatype;
  std::vector<int64_t> shape;
};

struct ModelMetadataPb {
  std::string name;
  std::vector<std::string> versions;
  std::string platform;
  std::vector<TensorMetadataPb> inputs;
  std::vector<TensorMetadataPb> outputs;
  static ModelMetadataPb Decode(const uint8_t* data, size_t n);
};

//==============================================================================
// ModelConfig — the COMPLETE message tree of model_config.proto
// (reference: src/rust/triton-client/proto/model_config.proto:86-2180).
// Every field of every submessage is decoded; unknown fields are
// skipped forward-compatibly.

struct ModelTensorReshapePb {
  std::vector<int64_t> shape;  // field 1
};

struct ModelTensorConfigPb {
  // shared shape of ModelInput (model_config.proto:317) and
  // ModelOutput (:428); per-side-only fields noted
  std::string name;
  int32_t data_type = 0;         // DataType enum value
  std::vector<int64_t> dims;
  ModelTensorReshapePb reshape;  // field 5 both sides
  bool has_reshape = false;
  std::string label_filename;    // outputs only (field 4)
  bool is_shape_tensor = false;
  bool is_non_linear_format_io = false;  // in:9 / out:7
  // inputs only:
  int32_t format = 0;            // Format enum (field 3)
  bool allow_ragged_batch = false;       // field 7
  bool optional_input = false;           // field 8
};

struct ModelRateLimiterPb {
  struct Resource {
    std::string name;
    bool global = false;
    uint32_t count = 0;
  };
  std::vector<Resource> resources;
  uint32_t priority = 0;
};

struct ModelInstanceGroupPb {
  struct SecondaryDevice {
    int32_t kind = 0;  // SecondaryDeviceKind
    int64_t device_id = 0;
  };
  std::string name;
  int32_t kind = 0;  // ModelInstanceGroup.Kind
  int32_t count = 0;
  ModelRateLimiterPb rate_limiter;  // field 6
  bool has_rate_limiter = false;
  std::vector<int32_t> gpus;
  std::vector<SecondaryDevice> secondary_devices;  // field 8
  std::vector<std::string> profile;                // field 5
  bool passive = false;                            // field 7
  std::string host_policy;                         // field 9
};

struct ModelVersionPolicyPb {
  // oneof policy_choice (model_config.proto:635)
  enum Choice { NONE, LATEST, ALL, SPECIFIC } choice = NONE;
  uint32_t latest_num_versions = 0;
  std::vector<int64_t> specific_versions;
};

struct ModelOptimizationPolicyPb {
  // model_config.proto:707
  struct Accelerator {
    std::string name;
    std::map<std::string, std::string> parameters;
  };
  struct GraphSpecShape {
    std::vector<int64_t> dim;
  };
  struct GraphSpec {
    int32_t batch_size = 0;
    std::map<std::string, GraphSpecShape> input;
    bool has_lower_bound = false;
    int32_t lower_bound_batch_size = 0;
    std::map<std::string, GraphSpecShape> lower_bound_input;
  };
  int32_t graph_level = 0;      // Graph.level
  bool has_graph = false;
  int32_t priority = 0;         // ModelPriority
  bool cuda_graphs = false;     // Cuda.graphs
  bool cuda_busy_wait_events = false;
  std::vector<GraphSpec> cuda_graph_spec;
  bool cuda_output_copy_stream = false;
  bool has_cuda = false;
  std::vector<Accelerator> gpu_execution_accelerator;
  std::vector<Accelerator> cpu_execution_accelerator;
  bool has_execution_accelerators = false;
  bool input_pinned_memory = false;   // PinnedMemoryBuffer.enable
  bool has_input_pinned_memory = false;
  bool output_pinned_memory = false;
  bool has_output_pinned_memory = false;
  uint32_t gather_kernel_buffer_threshold = 0;
  bool eager_batching = false;
};

struct ModelQueuePolicyPb {
  int32_t timeout_action = 0;  // REJECT=0 / DELAY=1
  uint64_t default_timeout_microseconds = 0;
  bool allow_timeout_override = false;
  uint32_t max_queue_size = 0;
};

struct ModelDynamicBatchingPb {
  std::vector<int32_t> preferred_batch_size;
  uint64_t max_queue_delay_microseconds = 0;
  bool preserve_ordering = false;
  uint64_t priority_levels = 0;
  uint64_t default_priority_level = 0;
  ModelQueuePolicyPb default_queue_policy;
  bool has_default_queue_policy = false;
  std::map<uint64_t, ModelQueuePolicyPb> priority_queue_policy;
};

struct ModelSequenceBatchingPb {
  // model_config.proto:1197
  struct Control {
    int32_t kind = 0;  // CONTROL_SEQUENCE_*
    std::vector<int32_t> int32_false_true;
    std::vector<float> fp32_false_true;
    std::vector<bool> bool_false_true;
    int32_t data_type = 0;
  };
  struct ControlInput {
    std::string name;
    std::vector<Control> control;
  };
  struct InitialState {
    int32_t data_type = 0;
    std::vector<int64_t> dims;
    enum DataChoice { NONE, ZERO, FILE } data_choice = NONE;
    bool zero_data = false;
    std::string data_file;
    std::string name;
  };
  struct State {
    std::string input_name;
    std::string output_name;
    int32_t data_type = 0;
    std::vector<int64_t> dims;
    std::vector<InitialState> initial_state;
    bool use_same_buffer_for_input_output = false;
    bool use_growable_memory = false;
  };
  enum Strategy { NONE, DIRECT, OLDEST } strategy = NONE;
  // direct
  uint64_t direct_max_queue_delay_microseconds = 0;
  float direct_minimum_slot_utilization = 0.f;
  // oldest
  int32_t oldest_max_candidate_sequences = 0;
  std::vector<int32_t> oldest_preferred_batch_size;
  uint64_t oldest_max_queue_delay_microseconds = 0;
  bool oldest_preserve_ordering = false;
  uint64_t max_sequence_idle_microseconds = 0;
  std::vector<ControlInput> control_input;
  std::vector<State> state;
  bool iterative_sequence = false;
};

struct EnsembleStepPb {
  std::string model_name;
  int64_t model_version = -1;
  std::map<std::string, std::string> input_map;
  std::map<std::string, std::string> output_map;
  std::string model_namespace;  // field 5
};

struct ModelWarmupPb {
  struct Input {
    int32_t data_type = 0;
    std::vector<int64_t> dims;
    enum DataChoice { NONE, ZERO, RANDOM, FILE } data_choice = NONE;
    bool zero_data = false;
    bool random_data = false;
    std::string input_data_file;
  };
  std::string name;
  uint32_t batch_size = 0;
  std::map<std::string, Input> inputs;
  uint32_t count = 0;
};

struct BatchInputPb {
  int32_t kind = 0;  // BatchInput.Kind
  std::vector<std::string> target_name;
  int32_t data_type = 0;
  std::vector<std::string> source_input;
};

struct BatchOutputPb {
  std::vector<std::string> target_name;
  int32_t kind = 0;  // BatchOutput.Kind
  std::vector<std::string> source_input;
};

struct ModelRepositoryAgentPb {
  std::string name;
  std::map<std::string, std::string> parameters;
};

struct ModelMetricControlPb {
  std::string family;              // metric_identifier.family
  std::vector<double> histogram_buckets;  // histogram_options.buckets
};

struct ModelConfigPb {
  std::string name;                          // 1
  std::string platform;                      // 2
  std::string backend;                       // 17
  std::string runtime;                       // 25
  ModelVersionPolicyPb version_policy;       // 3
  int32_t max_batch_size = 0;                // 4
  std::vector<ModelTensorConfigPb> input;    // 5
  std::vector<ModelTensorConfigPb> output;   // 6
  std::vector<BatchInputPb> batch_input;     // 20
  std::vector<BatchOutputPb> batch_output;   // 21
  ModelOptimizationPolicyPb optimization;    // 12
  bool has_optimization = false;
  // oneof scheduling_choice:
  bool has_dynamic_batching = false;         // 11
  bool has_sequence_batching = false;        // 13
  bool has_ensemble_scheduling = false;      // 15
  ModelDynamicBatchingPb dynamic_batching;
  ModelSequenceBatchingPb sequence_batching;
  std::vector<EnsembleStepPb> ensemble_steps;
  uint32_t ensemble_max_inflight_requests = 0;
  std::vector<ModelInstanceGroupPb> instance_group;       // 7
  std::string default_model_filename;                     // 8
  std::map<std::string, std::string> cc_model_filenames;  // 9
  std::map<std::string, std::string> metric_tags;         // 10
  std::map<std::string, std::string> parameters;  // 14, name->string_value
  std::vector<ModelWarmupPb> model_warmup;                // 16
  std::vector<std::string> op_library_filename;  // 18 ModelOperations
  bool decoupled = false;       // 19 model_transaction_policy
  std::vector<ModelRepositoryAgentPb> repository_agents;  // 23
  bool response_cache_enable = false;                     // 24
  std::vector<ModelMetricControlPb> metric_control;       // 26
  // Decodes a ModelConfigResponse (config = field 1).
  static ModelConfigPb Decode(const uint8_t* data, size_t n);
};

// Trace settings (grpc_service.proto:1673-1741): name -> list of
// string values, same map on request and response.
using TraceSettingsPb = std::map<std::string, std::vector<std::string>>;
std::string EncodeTraceSettingRequest(const TraceSettingsPb& settings,
                                      const std::string& model_name);
TraceSettingsPb DecodeTraceSettingResponse(const uint8_t* data, size_t n);

struct RepositoryIndexEntryPb {
  std::string name;
  std::string version;
  std::string state;
  std::string reason;
};
std::vector<RepositoryIndexEntryPb> DecodeRepositoryIndex(
    const uint8_t* data, size_t n);

// model_name = field 2 for load/unload requests
std::string EncodeRepositoryModelRequest(const std::string& model_name);

// Load with optional config-override JSON and file-override blobs
// (ModelRepositoryParameter string_param / bytes_param map)
std::string EncodeRepositoryModelLoadRequest(
    const std::string& model_name, const std::string& config,
    const std::map<std::string, std::string>& files);

// shm management
std::string EncodeSystemShmRegister(const std::string& name,
                                    const std::string& key, uint64_t offset,
                                    uint64_t byte_size);
std::string EncodeCudaShmRegister(const std::string& name,
                                  const std::string& raw_handle,
                                  int64_t device_id, uint64_t byte_size);

struct StatisticDurationPb {
  uint64_t count = 0;
  uint64_t ns = 0;
};

struct ModelStatisticsPb {
  std::string name;
  std::string version;
  uint64_t last_inference = 0;
  uint64_t inference_count = 0;
  uint64_t execution_count = 0;
  StatisticDurationPb success, fail, queue, compute_input, compute_infer,
      compute_output;
};
std::vector<ModelStatisticsPb> DecodeModelStatistics(const uint8_t* data,
                                                     size_t n);

}  // namespace kserve
}  // namespace client_amd
