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
// ModelConfig (model_config.proto:1971-2180). Decoded subset covering
// the commonly-consumed fields; unrecognized fields are skipped (the
// full schema lives on the Python side, client_amd/grpc/_proto.py).
struct ModelTensorConfigPb {
  std::string name;
  int32_t data_type = 0;  // DataType enum value
  std::vector<int64_t> dims;
  std::string label_filename;    // outputs only (field 4)
  bool is_shape_tensor = false;
  bool optional_input = false;   // inputs only (field 8)
};

struct ModelInstanceGroupPb {
  std::string name;
  int32_t kind = 0;  // ModelInstanceGroup.Kind
  int32_t count = 0;
  std::vector<int32_t> gpus;
};

struct ModelDynamicBatchingPb {
  std::vector<int32_t> preferred_batch_size;
  uint64_t max_queue_delay_microseconds = 0;
  bool preserve_ordering = false;
};

struct EnsembleStepPb {
  std::string model_name;
  int64_t model_version = -1;
  std::map<std::string, std::string> input_map;
  std::map<std::string, std::string> output_map;
};

struct ModelConfigPb {
  std::string name;
  std::string platform;
  std::string backend;
  std::string runtime;
  std::string default_model_filename;
  int32_t max_batch_size = 0;
  std::vector<ModelTensorConfigPb> input;
  std::vector<ModelTensorConfigPb> output;
  std::vector<ModelInstanceGroupPb> instance_group;
  bool has_dynamic_batching = false;   // scheduling_choice oneof
  bool has_sequence_batching = false;
  bool has_ensemble_scheduling = false;
  ModelDynamicBatchingPb dynamic_batching;
  std::vector<EnsembleStepPb> ensemble_steps;
  std::map<std::string, std::string> parameters;  // name -> string_value
  bool decoupled = false;              // model_transaction_policy
  bool response_cache_enable = false;
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
