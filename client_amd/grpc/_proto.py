"""KServe-v2 gRPC protobuf schema, constructed at runtime.

This environment has no protoc / grpcio-tools, so instead of generated
``*_pb2.py`` stubs the schema is authored here as a FileDescriptorProto
and turned into message classes via google.protobuf.message_factory.
Field numbers follow the KServe Predict Protocol v2 wire schema (the
reference vendors it at src/rust/triton-client/proto/grpc_service.proto;
e.g. ModelInferRequest :575-706 with raw_input_contents = 7,
ModelStreamInferResponse :821-840, shm messages :1419-1666, statistics
:881-1235, trace :1673-1741, log :1743-1780), so the bytes on the wire
are compatible with any KServe-v2 peer. ModelConfig is the full
subset of model_config.proto (full 2180-line schema: round 2).
"""

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_PKG = "inference"

# field type name -> FieldDescriptorProto type enum
_T = descriptor_pb2.FieldDescriptorProto
_TYPES = {
    "double": _T.TYPE_DOUBLE,
    "float": _T.TYPE_FLOAT,
    "int32": _T.TYPE_INT32,
    "int64": _T.TYPE_INT64,
    "uint32": _T.TYPE_UINT32,
    "uint64": _T.TYPE_UINT64,
    "bool": _T.TYPE_BOOL,
    "string": _T.TYPE_STRING,
    "bytes": _T.TYPE_BYTES,
}

# Schema DSL:
#   fields: (name, number, type[, flags])
#   type: scalar name | "msg:Fully.Qualified" | "enum:Name" |
#         "map:<keytype>:<valtype>"
#   flags: "r" repeated, "o<group>" oneof member
# Nested messages are written flat with dotted names (A.B).

_MESSAGES = {
    "ServerLiveRequest": [],
    "ServerLiveResponse": [("live", 1, "bool")],
    "ServerReadyRequest": [],
    "ServerReadyResponse": [("ready", 1, "bool")],
    "ModelReadyRequest": [("name", 1, "string"), ("version", 2, "string")],
    "ModelReadyResponse": [("ready", 1, "bool")],
    "ServerMetadataRequest": [],
    "ServerMetadataResponse": [
        ("name", 1, "string"),
        ("version", 2, "string"),
        ("extensions", 3, "string", "r"),
    ],
    "ModelMetadataRequest": [("name", 1, "string"), ("version", 2, "string")],
    "ModelMetadataResponse.TensorMetadata": [
        ("name", 1, "string"),
        ("datatype", 2, "string"),
        ("shape", 3, "int64", "r"),
    ],
    "ModelMetadataResponse": [
        ("name", 1, "string"),
        ("versions", 2, "string", "r"),
        ("platform", 3, "string"),
        ("inputs", 4, "msg:ModelMetadataResponse.TensorMetadata", "r"),
        ("outputs", 5, "msg:ModelMetadataResponse.TensorMetadata", "r"),
    ],
    "InferParameter": [
        ("bool_param", 1, "bool", "oparameter_choice"),
        ("int64_param", 2, "int64", "oparameter_choice"),
        ("string_param", 3, "string", "oparameter_choice"),
        ("double_param", 4, "double", "oparameter_choice"),
        ("uint64_param", 5, "uint64", "oparameter_choice"),
    ],
    "InferTensorContents": [
        ("bool_contents", 1, "bool", "r"),
        ("int_contents", 2, "int32", "r"),
        ("int64_contents", 3, "int64", "r"),
        ("uint_contents", 4, "uint32", "r"),
        ("uint64_contents", 5, "uint64", "r"),
        ("fp32_contents", 6, "float", "r"),
        ("fp64_contents", 7, "double", "r"),
        ("bytes_contents", 8, "bytes", "r"),
    ],
    "ModelInferRequest.InferInputTensor": [
        ("name", 1, "string"),
        ("datatype", 2, "string"),
        ("shape", 3, "int64", "r"),
        ("parameters", 4, "map:string:msg:InferParameter"),
        ("contents", 5, "msg:InferTensorContents"),
    ],
    "ModelInferRequest.InferRequestedOutputTensor": [
        ("name", 1, "string"),
        ("parameters", 2, "map:string:msg:InferParameter"),
    ],
    "ModelInferRequest": [
        ("model_name", 1, "string"),
        ("model_version", 2, "string"),
        ("id", 3, "string"),
        ("parameters", 4, "map:string:msg:InferParameter"),
        ("inputs", 5, "msg:ModelInferRequest.InferInputTensor", "r"),
        ("outputs", 6, "msg:ModelInferRequest.InferRequestedOutputTensor", "r"),
        ("raw_input_contents", 7, "bytes", "r"),
    ],
    "ModelInferResponse.InferOutputTensor": [
        ("name", 1, "string"),
        ("datatype", 2, "string"),
        ("shape", 3, "int64", "r"),
        ("parameters", 4, "map:string:msg:InferParameter"),
        ("contents", 5, "msg:InferTensorContents"),
    ],
    "ModelInferResponse": [
        ("model_name", 1, "string"),
        ("model_version", 2, "string"),
        ("id", 3, "string"),
        ("parameters", 4, "map:string:msg:InferParameter"),
        ("outputs", 5, "msg:ModelInferResponse.InferOutputTensor", "r"),
        ("raw_output_contents", 6, "bytes", "r"),
    ],
    "ModelStreamInferResponse": [
        ("error_message", 1, "string"),
        ("infer_response", 2, "msg:ModelInferResponse"),
    ],
    # ---- model config (full reference schema, model_config.proto:86-2180) ----
    "ModelRateLimiter.Resource": [
        ("name", 1, "string"),
        ("global", 2, "bool"),
        ("count", 3, "uint32"),
    ],
    "ModelRateLimiter": [
        ("resources", 1, "msg:ModelRateLimiter.Resource", "r"),
        ("priority", 2, "uint32"),
    ],
    "ModelInstanceGroup.SecondaryDevice": [
        ("kind", 1, "enum:ModelInstanceGroup.SecondaryDevice.SecondaryDeviceKind"),
        ("device_id", 2, "int64"),
    ],
    "ModelInstanceGroup": [
        ("name", 1, "string"),
        ("kind", 4, "enum:ModelInstanceGroup.Kind"),
        ("count", 2, "int32"),
        ("rate_limiter", 6, "msg:ModelRateLimiter"),
        ("gpus", 3, "int32", "r"),
        ("secondary_devices", 8, "msg:ModelInstanceGroup.SecondaryDevice", "r"),
        ("profile", 5, "string", "r"),
        ("passive", 7, "bool"),
        ("host_policy", 9, "string"),
    ],
    "ModelTensorReshape": [("shape", 1, "int64", "r")],
    "ModelInput": [
        ("name", 1, "string"),
        ("data_type", 2, "enum:DataType"),
        ("format", 3, "enum:ModelInput.Format"),
        ("dims", 4, "int64", "r"),
        ("reshape", 5, "msg:ModelTensorReshape"),
        ("is_shape_tensor", 6, "bool"),
        ("allow_ragged_batch", 7, "bool"),
        ("optional", 8, "bool"),
        ("is_non_linear_format_io", 9, "bool"),
    ],
    "ModelOutput": [
        ("name", 1, "string"),
        ("data_type", 2, "enum:DataType"),
        ("dims", 3, "int64", "r"),
        ("label_filename", 4, "string"),
        ("reshape", 5, "msg:ModelTensorReshape"),
        ("is_shape_tensor", 6, "bool"),
        ("is_non_linear_format_io", 7, "bool"),
    ],
    "BatchInput": [
        ("kind", 1, "enum:BatchInput.Kind"),
        ("target_name", 2, "string", "r"),
        ("data_type", 3, "enum:DataType"),
        ("source_input", 4, "string", "r"),
    ],
    "BatchOutput": [
        ("target_name", 1, "string", "r"),
        ("kind", 2, "enum:BatchOutput.Kind"),
        ("source_input", 3, "string", "r"),
    ],
    "ModelVersionPolicy.Latest": [("num_versions", 1, "uint32")],
    "ModelVersionPolicy.All": [],
    "ModelVersionPolicy.Specific": [("versions", 1, "int64", "r")],
    "ModelVersionPolicy": [
        ("latest", 1, "msg:ModelVersionPolicy.Latest", "opolicy_choice"),
        ("all", 2, "msg:ModelVersionPolicy.All", "opolicy_choice"),
        ("specific", 3, "msg:ModelVersionPolicy.Specific", "opolicy_choice"),
    ],
    "ModelOptimizationPolicy.Graph": [("level", 1, "int32")],
    "ModelOptimizationPolicy.Cuda.GraphSpec.Shape": [("dim", 1, "int64", "r")],
    "ModelOptimizationPolicy.Cuda.GraphSpec.LowerBound": [
        ("batch_size", 1, "int32"),
        ("input", 2, "map:string:msg:ModelOptimizationPolicy.Cuda.GraphSpec.Shape"),
    ],
    "ModelOptimizationPolicy.Cuda.GraphSpec": [
        ("batch_size", 1, "int32"),
        ("input", 2, "map:string:msg:ModelOptimizationPolicy.Cuda.GraphSpec.Shape"),
        ("graph_lower_bound", 3, "msg:ModelOptimizationPolicy.Cuda.GraphSpec.LowerBound"),
    ],
    "ModelOptimizationPolicy.Cuda": [
        ("graphs", 1, "bool"),
        ("busy_wait_events", 2, "bool"),
        ("graph_spec", 3, "msg:ModelOptimizationPolicy.Cuda.GraphSpec", "r"),
        ("output_copy_stream", 4, "bool"),
    ],
    "ModelOptimizationPolicy.ExecutionAccelerators.Accelerator": [
        ("name", 1, "string"),
        ("parameters", 2, "map:string:string"),
    ],
    "ModelOptimizationPolicy.ExecutionAccelerators": [
        ("gpu_execution_accelerator", 1,
         "msg:ModelOptimizationPolicy.ExecutionAccelerators.Accelerator", "r"),
        ("cpu_execution_accelerator", 2,
         "msg:ModelOptimizationPolicy.ExecutionAccelerators.Accelerator", "r"),
    ],
    "ModelOptimizationPolicy.PinnedMemoryBuffer": [("enable", 1, "bool")],
    "ModelOptimizationPolicy": [
        ("graph", 1, "msg:ModelOptimizationPolicy.Graph"),
        ("priority", 2, "enum:ModelOptimizationPolicy.ModelPriority"),
        ("cuda", 3, "msg:ModelOptimizationPolicy.Cuda"),
        ("execution_accelerators", 4,
         "msg:ModelOptimizationPolicy.ExecutionAccelerators"),
        ("input_pinned_memory", 5,
         "msg:ModelOptimizationPolicy.PinnedMemoryBuffer"),
        ("output_pinned_memory", 6,
         "msg:ModelOptimizationPolicy.PinnedMemoryBuffer"),
        ("gather_kernel_buffer_threshold", 7, "uint32"),
        ("eager_batching", 8, "bool"),
    ],
    "ModelQueuePolicy": [
        ("timeout_action", 1, "enum:ModelQueuePolicy.TimeoutAction"),
        ("default_timeout_microseconds", 2, "uint64"),
        ("allow_timeout_override", 3, "bool"),
        ("max_queue_size", 4, "uint32"),
    ],
    "ModelDynamicBatching": [
        ("preferred_batch_size", 1, "int32", "r"),
        ("max_queue_delay_microseconds", 2, "uint64"),
        ("preserve_ordering", 3, "bool"),
        ("priority_levels", 4, "uint64"),
        ("default_priority_level", 5, "uint64"),
        ("default_queue_policy", 6, "msg:ModelQueuePolicy"),
        ("priority_queue_policy", 7, "map:uint64:msg:ModelQueuePolicy"),
    ],
    "ModelSequenceBatching.Control": [
        ("kind", 1, "enum:ModelSequenceBatching.Control.Kind"),
        ("int32_false_true", 2, "int32", "r"),
        ("fp32_false_true", 3, "float", "r"),
        ("bool_false_true", 5, "bool", "r"),
        ("data_type", 4, "enum:DataType"),
    ],
    "ModelSequenceBatching.ControlInput": [
        ("name", 1, "string"),
        ("control", 2, "msg:ModelSequenceBatching.Control", "r"),
    ],
    "ModelSequenceBatching.InitialState": [
        ("data_type", 1, "enum:DataType"),
        ("dims", 2, "int64", "r"),
        ("zero_data", 3, "bool", "ostate_data"),
        ("data_file", 4, "string", "ostate_data"),
        ("name", 5, "string"),
    ],
    "ModelSequenceBatching.State": [
        ("input_name", 1, "string"),
        ("output_name", 2, "string"),
        ("data_type", 3, "enum:DataType"),
        ("dims", 4, "int64", "r"),
        ("initial_state", 5, "msg:ModelSequenceBatching.InitialState", "r"),
        ("use_same_buffer_for_input_output", 6, "bool"),
        ("use_growable_memory", 7, "bool"),
    ],
    "ModelSequenceBatching.StrategyDirect": [
        ("max_queue_delay_microseconds", 1, "uint64"),
        ("minimum_slot_utilization", 2, "float"),
    ],
    "ModelSequenceBatching.StrategyOldest": [
        ("max_candidate_sequences", 1, "int32"),
        ("preferred_batch_size", 2, "int32", "r"),
        ("max_queue_delay_microseconds", 3, "uint64"),
        ("preserve_ordering", 4, "bool"),
    ],
    "ModelSequenceBatching": [
        ("direct", 3, "msg:ModelSequenceBatching.StrategyDirect",
         "ostrategy_choice"),
        ("oldest", 4, "msg:ModelSequenceBatching.StrategyOldest",
         "ostrategy_choice"),
        ("max_sequence_idle_microseconds", 1, "uint64"),
        ("control_input", 2, "msg:ModelSequenceBatching.ControlInput", "r"),
        ("state", 5, "msg:ModelSequenceBatching.State", "r"),
        ("iterative_sequence", 6, "bool"),
    ],
    "ModelEnsembling.Step": [
        ("model_name", 1, "string"),
        ("model_version", 2, "int64"),
        ("input_map", 3, "map:string:string"),
        ("output_map", 4, "map:string:string"),
        ("model_namespace", 5, "string"),
    ],
    "ModelEnsembling": [
        ("step", 1, "msg:ModelEnsembling.Step", "r"),
        ("max_inflight_requests", 2, "uint32"),
    ],
    "ModelParameter": [("string_value", 1, "string")],
    "ModelWarmup.Input": [
        ("data_type", 1, "enum:DataType"),
        ("dims", 2, "int64", "r"),
        ("zero_data", 3, "bool", "oinput_data_type"),
        ("random_data", 4, "bool", "oinput_data_type"),
        ("input_data_file", 5, "string", "oinput_data_type"),
    ],
    "ModelWarmup": [
        ("name", 1, "string"),
        ("batch_size", 2, "uint32"),
        ("inputs", 3, "map:string:msg:ModelWarmup.Input"),
        ("count", 4, "uint32"),
    ],
    "ModelOperations": [("op_library_filename", 1, "string", "r")],
    "ModelTransactionPolicy": [("decoupled", 1, "bool")],
    "ModelRepositoryAgents.Agent": [
        ("name", 1, "string"),
        ("parameters", 2, "map:string:string"),
    ],
    "ModelRepositoryAgents": [
        ("agents", 1, "msg:ModelRepositoryAgents.Agent", "r"),
    ],
    "ModelResponseCache": [("enable", 1, "bool")],
    "ModelMetrics.MetricControl.MetricIdentifier": [("family", 1, "string")],
    "ModelMetrics.MetricControl.HistogramOptions": [
        ("buckets", 1, "double", "r"),
    ],
    "ModelMetrics.MetricControl": [
        ("metric_identifier", 1,
         "msg:ModelMetrics.MetricControl.MetricIdentifier"),
        ("histogram_options", 2,
         "msg:ModelMetrics.MetricControl.HistogramOptions",
         "ometric_options"),
    ],
    "ModelMetrics": [
        ("metric_control", 1, "msg:ModelMetrics.MetricControl", "r"),
    ],
    "ModelConfig": [
        ("name", 1, "string"),
        ("platform", 2, "string"),
        ("backend", 17, "string"),
        ("runtime", 25, "string"),
        ("version_policy", 3, "msg:ModelVersionPolicy"),
        ("max_batch_size", 4, "int32"),
        ("input", 5, "msg:ModelInput", "r"),
        ("output", 6, "msg:ModelOutput", "r"),
        ("batch_input", 20, "msg:BatchInput", "r"),
        ("batch_output", 21, "msg:BatchOutput", "r"),
        ("optimization", 12, "msg:ModelOptimizationPolicy"),
        ("dynamic_batching", 11, "msg:ModelDynamicBatching",
         "oscheduling_choice"),
        ("sequence_batching", 13, "msg:ModelSequenceBatching",
         "oscheduling_choice"),
        ("ensemble_scheduling", 15, "msg:ModelEnsembling",
         "oscheduling_choice"),
        ("instance_group", 7, "msg:ModelInstanceGroup", "r"),
        ("default_model_filename", 8, "string"),
        ("cc_model_filenames", 9, "map:string:string"),
        ("metric_tags", 10, "map:string:string"),
        ("parameters", 14, "map:string:msg:ModelParameter"),
        ("model_warmup", 16, "msg:ModelWarmup", "r"),
        ("model_operations", 18, "msg:ModelOperations"),
        ("model_transaction_policy", 19, "msg:ModelTransactionPolicy"),
        ("model_repository_agents", 23, "msg:ModelRepositoryAgents"),
        ("response_cache", 24, "msg:ModelResponseCache"),
        ("model_metrics", 26, "msg:ModelMetrics"),
    ],
    "ModelConfigRequest": [("name", 1, "string"), ("version", 2, "string")],
    "ModelConfigResponse": [("config", 1, "msg:ModelConfig")],
    # ---- repository ----
    "RepositoryIndexRequest": [
        ("repository_name", 1, "string"),
        ("ready", 2, "bool"),
    ],
    "RepositoryIndexResponse.ModelIndex": [
        ("name", 1, "string"),
        ("version", 2, "string"),
        ("state", 3, "string"),
        ("reason", 4, "string"),
    ],
    "RepositoryIndexResponse": [
        ("models", 1, "msg:RepositoryIndexResponse.ModelIndex", "r"),
    ],
    "ModelRepositoryParameter": [
        ("bool_param", 1, "bool", "oparameter_choice"),
        ("int64_param", 2, "int64", "oparameter_choice"),
        ("string_param", 3, "string", "oparameter_choice"),
        ("bytes_param", 4, "bytes", "oparameter_choice"),
    ],
    "RepositoryModelLoadRequest": [
        ("repository_name", 1, "string"),
        ("model_name", 2, "string"),
        ("parameters", 3, "map:string:msg:ModelRepositoryParameter"),
    ],
    "RepositoryModelLoadResponse": [],
    "RepositoryModelUnloadRequest": [
        ("repository_name", 1, "string"),
        ("model_name", 2, "string"),
        ("parameters", 3, "map:string:msg:ModelRepositoryParameter"),
    ],
    "RepositoryModelUnloadResponse": [],
    # ---- shared memory ----
    "SystemSharedMemoryStatusRequest": [("name", 1, "string")],
    "SystemSharedMemoryStatusResponse.RegionStatus": [
        ("name", 1, "string"),
        ("key", 2, "string"),
        ("offset", 3, "uint64"),
        ("byte_size", 4, "uint64"),
    ],
    "SystemSharedMemoryStatusResponse": [
        ("regions", 1, "map:string:msg:SystemSharedMemoryStatusResponse.RegionStatus"),
    ],
    "SystemSharedMemoryRegisterRequest": [
        ("name", 1, "string"),
        ("key", 2, "string"),
        ("offset", 3, "uint64"),
        ("byte_size", 4, "uint64"),
    ],
    "SystemSharedMemoryRegisterResponse": [],
    "SystemSharedMemoryUnregisterRequest": [("name", 1, "string")],
    "SystemSharedMemoryUnregisterResponse": [],
    "CudaSharedMemoryStatusRequest": [("name", 1, "string")],
    "CudaSharedMemoryStatusResponse.RegionStatus": [
        ("name", 1, "string"),
        ("device_id", 2, "uint64"),
        ("byte_size", 3, "uint64"),
    ],
    "CudaSharedMemoryStatusResponse": [
        ("regions", 1, "map:string:msg:CudaSharedMemoryStatusResponse.RegionStatus"),
    ],
    "CudaSharedMemoryRegisterRequest": [
        ("name", 1, "string"),
        ("raw_handle", 2, "bytes"),
        ("device_id", 3, "int64"),
        ("byte_size", 4, "uint64"),
    ],
    "CudaSharedMemoryRegisterResponse": [],
    "CudaSharedMemoryUnregisterRequest": [("name", 1, "string")],
    "CudaSharedMemoryUnregisterResponse": [],
    # ---- statistics ----
    "ModelStatisticsRequest": [("name", 1, "string"), ("version", 2, "string")],
    "StatisticDuration": [("count", 1, "uint64"), ("ns", 2, "uint64")],
    "InferStatistics": [
        ("success", 1, "msg:StatisticDuration"),
        ("fail", 2, "msg:StatisticDuration"),
        ("queue", 3, "msg:StatisticDuration"),
        ("compute_input", 4, "msg:StatisticDuration"),
        ("compute_infer", 5, "msg:StatisticDuration"),
        ("compute_output", 6, "msg:StatisticDuration"),
        ("cache_hit", 7, "msg:StatisticDuration"),
        ("cache_miss", 8, "msg:StatisticDuration"),
    ],
    "InferBatchStatistics": [
        ("batch_size", 1, "uint64"),
        ("compute_input", 2, "msg:StatisticDuration"),
        ("compute_infer", 3, "msg:StatisticDuration"),
        ("compute_output", 4, "msg:StatisticDuration"),
    ],
    "MemoryUsage": [
        ("type", 1, "string"),
        ("id", 2, "int64"),
        ("byte_size", 3, "uint64"),
    ],
    "ModelStatistics": [
        ("name", 1, "string"),
        ("version", 2, "string"),
        ("last_inference", 3, "uint64"),
        ("inference_count", 4, "uint64"),
        ("execution_count", 5, "uint64"),
        ("inference_stats", 6, "msg:InferStatistics"),
        ("batch_stats", 7, "msg:InferBatchStatistics", "r"),
        ("memory_usage", 8, "msg:MemoryUsage", "r"),
    ],
    "ModelStatisticsResponse": [
        ("model_stats", 1, "msg:ModelStatistics", "r"),
    ],
    # ---- trace / log settings ----
    "TraceSettingRequest.SettingValue": [("value", 1, "string", "r")],
    "TraceSettingRequest": [
        ("settings", 1, "map:string:msg:TraceSettingRequest.SettingValue"),
        ("model_name", 2, "string"),
    ],
    "TraceSettingResponse.SettingValue": [("value", 1, "string", "r")],
    "TraceSettingResponse": [
        ("settings", 1, "map:string:msg:TraceSettingResponse.SettingValue"),
    ],
    "LogSettingsRequest.SettingValue": [
        ("bool_param", 1, "bool", "oparameter_choice"),
        ("uint32_param", 2, "uint32", "oparameter_choice"),
        ("string_param", 3, "string", "oparameter_choice"),
    ],
    "LogSettingsRequest": [
        ("settings", 1, "map:string:msg:LogSettingsRequest.SettingValue"),
        ("log_file", 2, "string"),
    ],
    "LogSettingsResponse.SettingValue": [
        ("bool_param", 1, "bool", "oparameter_choice"),
        ("uint32_param", 2, "uint32", "oparameter_choice"),
        ("string_param", 3, "string", "oparameter_choice"),
    ],
    "LogSettingsResponse": [
        ("settings", 1, "map:string:msg:LogSettingsResponse.SettingValue"),
    ],
}

_ENUMS = {
    "DataType": [
        "TYPE_INVALID", "TYPE_BOOL", "TYPE_UINT8", "TYPE_UINT16", "TYPE_UINT32",
        "TYPE_UINT64", "TYPE_INT8", "TYPE_INT16", "TYPE_INT32", "TYPE_INT64",
        "TYPE_FP16", "TYPE_FP32", "TYPE_FP64", "TYPE_STRING", "TYPE_BF16",
    ],
    "ModelInput.Format": ["FORMAT_NONE", "FORMAT_NHWC", "FORMAT_NCHW"],
    "ModelInstanceGroup.Kind": [
        "KIND_AUTO", "KIND_GPU", "KIND_CPU", "KIND_MODEL",
    ],
    "ModelInstanceGroup.SecondaryDevice.SecondaryDeviceKind": ["KIND_NVDLA"],
    "BatchInput.Kind": [
        "BATCH_ELEMENT_COUNT", "BATCH_ACCUMULATED_ELEMENT_COUNT",
        "BATCH_ACCUMULATED_ELEMENT_COUNT_WITH_ZERO",
        "BATCH_MAX_ELEMENT_COUNT_AS_SHAPE", "BATCH_ITEM_SHAPE",
        "BATCH_ITEM_SHAPE_FLATTEN",
    ],
    "BatchOutput.Kind": ["BATCH_SCATTER_WITH_INPUT_SHAPE"],
    "ModelOptimizationPolicy.ModelPriority": [
        "PRIORITY_DEFAULT", "PRIORITY_MAX", "PRIORITY_MIN",
    ],
    "ModelQueuePolicy.TimeoutAction": ["REJECT", "DELAY"],
    "ModelSequenceBatching.Control.Kind": [
        "CONTROL_SEQUENCE_START", "CONTROL_SEQUENCE_READY",
        "CONTROL_SEQUENCE_END", "CONTROL_SEQUENCE_CORRID",
    ],
}


def _build_file_descriptor():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "client_amd/grpc_service.proto"
    fdp.package = _PKG
    fdp.syntax = "proto3"

    # Top-level containers for nested message/enum lookup.
    top_msgs = {}

    def get_container(path):
        """Return (messages_list, enums_list) for a dotted parent path."""
        if not path:
            return fdp.message_type, fdp.enum_type
        m = top_msgs[path]
        return m.nested_type, m.enum_type

    # Create message shells first (parents before children).
    for full_name in sorted(_MESSAGES, key=lambda n: n.count(".")):
        parent, _, short = full_name.rpartition(".")
        msgs, _ = get_container(parent)
        m = msgs.add()
        m.name = short
        top_msgs[full_name] = m

    # Enums.
    for full_name, values in _ENUMS.items():
        parent, _, short = full_name.rpartition(".")
        if parent and parent not in top_msgs:
            # parent message must exist (ModelInput etc.)
            raise RuntimeError(f"enum parent {parent} missing")
        _, enums = get_container(parent)
        e = enums.add()
        e.name = short
        for i, v in enumerate(values):
            ev = e.value.add()
            ev.name = v
            ev.number = i

    # Fields.
    for full_name, fields in _MESSAGES.items():
        m = top_msgs[full_name]
        oneofs = {}
        for spec in fields:
            name, number, ftype = spec[0], spec[1], spec[2]
            flags = spec[3] if len(spec) > 3 else ""
            f = m.field.add()
            f.name = name
            f.number = number
            if ftype.startswith("map:"):
                # map:<keytype>:<valspec>  (valspec may itself contain ':')
                _, keytype, valspec = ftype.split(":", 2)
                entry = m.nested_type.add()
                entry.name = _map_entry_name(name)
                entry.options.map_entry = True
                kf = entry.field.add()
                kf.name = "key"
                kf.number = 1
                kf.label = _T.LABEL_OPTIONAL
                kf.type = _TYPES[keytype]
                vf = entry.field.add()
                vf.name = "value"
                vf.number = 2
                vf.label = _T.LABEL_OPTIONAL
                _set_type(vf, valspec)
                f.label = _T.LABEL_REPEATED
                f.type = _T.TYPE_MESSAGE
                f.type_name = f".{_PKG}.{full_name}.{entry.name}"
            else:
                f.label = _T.LABEL_REPEATED if "r" in flags and not flags.startswith("o") else _T.LABEL_OPTIONAL
                _set_type(f, ftype)
                if flags.startswith("o"):
                    group = flags[1:]
                    if group not in oneofs:
                        od = m.oneof_decl.add()
                        od.name = group
                        oneofs[group] = len(m.oneof_decl) - 1
                    f.oneof_index = oneofs[group]
    return fdp


def _map_entry_name(field_name):
    return "".join(p.capitalize() for p in field_name.split("_")) + "Entry"


def _set_type(f, ftype):
    if ftype.startswith("msg:"):
        f.type = _T.TYPE_MESSAGE
        f.type_name = f".{_PKG}." + ftype[4:]
    elif ftype.startswith("enum:"):
        f.type = _T.TYPE_ENUM
        f.type_name = f".{_PKG}." + ftype[5:]
    else:
        f.type = _TYPES[ftype]


_pool = descriptor_pool.DescriptorPool()
_pool.Add(_build_file_descriptor())


class _Namespace:
    pass


def _load():
    ns = _Namespace()
    # Parents before children so nested classes can attach to them.
    for full_name in sorted(_MESSAGES, key=lambda n: n.count(".")):
        cls = message_factory.GetMessageClass(
            _pool.FindMessageTypeByName(f"{_PKG}.{full_name}")
        )
        # Expose nested classes under their parent, like generated pb2 code.
        parent, _, short = full_name.rpartition(".")
        if parent:
            setattr(getattr(ns, parent.replace(".", "_")), short, cls)
            setattr(ns, full_name.replace(".", "_"), cls)
        else:
            setattr(ns, full_name, cls)
    for full_name in _ENUMS:
        try:
            ed = _pool.FindEnumTypeByName(f"{_PKG}.{full_name}")
            if "." not in full_name:
                setattr(ns, full_name, ed)
        except KeyError:
            pass
    return ns


service_pb2 = _load()

SERVICE_NAME = "inference.GRPCInferenceService"

# RPC name -> (request class, response class, streaming)
RPCS = {
    "ServerLive": (service_pb2.ServerLiveRequest, service_pb2.ServerLiveResponse, False),
    "ServerReady": (service_pb2.ServerReadyRequest, service_pb2.ServerReadyResponse, False),
    "ModelReady": (service_pb2.ModelReadyRequest, service_pb2.ModelReadyResponse, False),
    "ServerMetadata": (service_pb2.ServerMetadataRequest, service_pb2.ServerMetadataResponse, False),
    "ModelMetadata": (service_pb2.ModelMetadataRequest, service_pb2.ModelMetadataResponse, False),
    "ModelInfer": (service_pb2.ModelInferRequest, service_pb2.ModelInferResponse, False),
    "ModelStreamInfer": (service_pb2.ModelInferRequest, service_pb2.ModelStreamInferResponse, True),
    "ModelConfig": (service_pb2.ModelConfigRequest, service_pb2.ModelConfigResponse, False),
    "ModelStatistics": (service_pb2.ModelStatisticsRequest, service_pb2.ModelStatisticsResponse, False),
    "RepositoryIndex": (service_pb2.RepositoryIndexRequest, service_pb2.RepositoryIndexResponse, False),
    "RepositoryModelLoad": (service_pb2.RepositoryModelLoadRequest, service_pb2.RepositoryModelLoadResponse, False),
    "RepositoryModelUnload": (service_pb2.RepositoryModelUnloadRequest, service_pb2.RepositoryModelUnloadResponse, False),
    "SystemSharedMemoryStatus": (service_pb2.SystemSharedMemoryStatusRequest, service_pb2.SystemSharedMemoryStatusResponse, False),
    "SystemSharedMemoryRegister": (service_pb2.SystemSharedMemoryRegisterRequest, service_pb2.SystemSharedMemoryRegisterResponse, False),
    "SystemSharedMemoryUnregister": (service_pb2.SystemSharedMemoryUnregisterRequest, service_pb2.SystemSharedMemoryUnregisterResponse, False),
    "CudaSharedMemoryStatus": (service_pb2.CudaSharedMemoryStatusRequest, service_pb2.CudaSharedMemoryStatusResponse, False),
    "CudaSharedMemoryRegister": (service_pb2.CudaSharedMemoryRegisterRequest, service_pb2.CudaSharedMemoryRegisterResponse, False),
    "CudaSharedMemoryUnregister": (service_pb2.CudaSharedMemoryUnregisterRequest, service_pb2.CudaSharedMemoryUnregisterResponse, False),
    "TraceSetting": (service_pb2.TraceSettingRequest, service_pb2.TraceSettingResponse, False),
    "LogSettings": (service_pb2.LogSettingsRequest, service_pb2.LogSettingsResponse, False),
}
