"""Wire-compatible protobuf messages for the reference engine protocol.

The reference gateway speaks `sglang.grpc.scheduler.SglangScheduler`
(/root/reference/crates/grpc_client/proto/sglang_scheduler.proto:11-59) and
the shared `smg.grpc.common` types (common.proto) to its engine servicers.
This image has the protobuf RUNTIME but no protoc/grpcio-tools, so the same
message schema (field names, numbers, types, oneofs, maps, proto3 optionals
— transcribed from the reference protos) is built here programmatically as a
FileDescriptorProto and turned into message classes via message_factory.

Anything that serializes with these classes is byte-compatible with the
reference's tonic clients / grpcio servicers; tests/test_proto_wire.py proves
the encoding against a first-principles wire-format encoder.

The msgpack surface (grpc/api.py) remains the in-repo default; these messages
are the interop path for engines/routers that speak the reference protocol.
"""
from __future__ import annotations

from typing import Dict

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory
from google.protobuf import struct_pb2, timestamp_pb2

F = descriptor_pb2.FieldDescriptorProto

# scalar type shorthands
T_STR = F.TYPE_STRING
T_U32 = F.TYPE_UINT32
T_U64 = F.TYPE_UINT64
T_I32 = F.TYPE_INT32
T_I64 = F.TYPE_INT64
T_F32 = F.TYPE_FLOAT
T_F64 = F.TYPE_DOUBLE
T_BOOL = F.TYPE_BOOL
T_BYTES = F.TYPE_BYTES
T_MSG = F.TYPE_MESSAGE
T_ENUM = F.TYPE_ENUM


class _Msg:
    def __init__(self, proto: descriptor_pb2.DescriptorProto):
        self.p = proto
        self._optionals = []  # fields needing synthetic proto3-optional oneofs

    def field(self, name, num, ftype, *, repeated=False, type_name=None,
              optional=False, oneof=None):
        f = self.p.field.add()
        f.name = name
        f.number = num
        f.type = ftype
        f.label = F.LABEL_REPEATED if repeated else F.LABEL_OPTIONAL
        if type_name:
            f.type_name = type_name
        if oneof is not None:
            f.oneof_index = oneof
        if optional:
            f.proto3_optional = True
            self._optionals.append(f)
        return self

    def oneof(self, name) -> int:
        o = self.p.oneof_decl.add()
        o.name = name
        return len(self.p.oneof_decl) - 1

    def map_field(self, name, num, key_type, val_type, *, val_type_name=None):
        # a map<K,V> is a repeated nested MapEntry message
        entry = self.p.nested_type.add()
        entry.name = "".join(w.capitalize() for w in name.split("_")) + "Entry"
        entry.options.map_entry = True
        k = entry.field.add(); k.name = "key"; k.number = 1; k.type = key_type
        k.label = F.LABEL_OPTIONAL
        v = entry.field.add(); v.name = "value"; v.number = 2; v.type = val_type
        v.label = F.LABEL_OPTIONAL
        if val_type_name:
            v.type_name = val_type_name
        self.field(name, num, T_MSG, repeated=True,
                   type_name=f".{_CUR_PKG[0]}.{self.p.name}.{entry.name}")
        return self

    def finish(self):
        # synthetic oneofs for proto3 optional fields go after real oneofs
        for f in self._optionals:
            o = self.p.oneof_decl.add()
            o.name = "_" + f.name
            f.oneof_index = len(self.p.oneof_decl) - 1
        self._optionals = []


_CUR_PKG = [""]


class _File:
    def __init__(self, name, package, deps=()):
        self.fd = descriptor_pb2.FileDescriptorProto()
        self.fd.name = name
        self.fd.package = package
        self.fd.syntax = "proto3"
        self.fd.dependency.extend(deps)
        self.msgs = []
        _CUR_PKG[0] = package

    def message(self, name) -> _Msg:
        _CUR_PKG[0] = self.fd.package
        m = _Msg(self.fd.message_type.add())
        m.p.name = name
        self.msgs.append(m)
        return m

    def enum(self, name, values):
        e = self.fd.enum_type.add()
        e.name = name
        for vname, vnum in values:
            v = e.value.add()
            v.name = vname
            v.number = vnum

    def build(self, pool):
        for m in self.msgs:
            m.finish()
        pool.Add(self.fd)


_pool = descriptor_pool.DescriptorPool()
# well-known dependencies into the private pool
for _wkt in (struct_pb2, timestamp_pb2):
    _fd = descriptor_pb2.FileDescriptorProto.FromString(_wkt.DESCRIPTOR.serialized_pb)
    _pool.Add(_fd)

C = "smg.grpc.common"
S = "sglang.grpc.scheduler"

# --------------------------------------------------------------------------
# common.proto (reference crates/grpc_client/proto/common.proto)
# --------------------------------------------------------------------------
_cf = _File("common.proto", C)
_cf.message("GetTokenizerRequest")
_cf.message("GetTokenizerChunk") \
    .field("data", 1, T_BYTES).field("sha256", 2, T_STR)
_cf.message("SubscribeKvEventsRequest").field("start_sequence_number", 1, T_U64)
_cf.message("KvEventBatch") \
    .field("sequence_number", 1, T_U64).field("timestamp", 2, T_F64) \
    .field("events", 3, T_MSG, repeated=True, type_name=f".{C}.KvCacheEvent") \
    .field("dp_rank", 4, T_I32, optional=True)
_m = _cf.message("KvCacheEvent")
_m.field("event_id", 1, T_U64)
_o = _m.oneof("data")
_m.field("stored", 2, T_MSG, type_name=f".{C}.KvBlocksStored", oneof=_o)
_m.field("removed", 3, T_MSG, type_name=f".{C}.KvBlocksRemoved", oneof=_o)
_m.field("cleared", 4, T_MSG, type_name=f".{C}.KvCacheCleared", oneof=_o)
_cf.message("KvBlocksStored") \
    .field("blocks", 1, T_MSG, repeated=True, type_name=f".{C}.KvBlock") \
    .field("parent_block_hash", 2, T_I64, optional=True)
_cf.message("KvBlock") \
    .field("block_hash", 1, T_I64).field("token_ids", 2, T_U32, repeated=True) \
    .field("block_size", 3, T_I32).field("lora_id", 4, T_I64, optional=True) \
    .field("cache_level", 5, T_I32, optional=True)
_cf.message("KvBlocksRemoved") \
    .field("block_hashes", 1, T_I64, repeated=True) \
    .field("cache_level", 2, T_I32, optional=True)
_cf.message("KvCacheCleared")
_cf.message("FlushCacheRequest").field("timeout_s", 1, T_F32)
_cf.message("FlushCacheResponse").field("success", 1, T_BOOL).field("message", 2, T_STR)
_cf.message("StartProfileRequest") \
    .field("output_dir", 1, T_STR, optional=True) \
    .field("start_step", 2, T_I32, optional=True) \
    .field("num_steps", 3, T_I32, optional=True) \
    .field("activities", 4, T_STR, repeated=True) \
    .field("with_stack", 5, T_BOOL, optional=True) \
    .field("record_shapes", 6, T_BOOL, optional=True) \
    .field("profile_by_stage", 7, T_BOOL)
_cf.message("StopProfileRequest")
_cf.message("ProfileResponse").field("success", 1, T_BOOL).field("message", 2, T_STR)
_cf.message("ShmHandle") \
    .field("name", 1, T_STR).field("offset", 2, T_U64).field("nbytes", 3, T_U64) \
    .field("owner_id", 4, T_STR)
_cf.message("RemoteTensorHandle") \
    .field("transport", 1, T_STR).field("descriptor", 2, T_BYTES).field("nbytes", 3, T_U64)
_cf.enum("Modality", [("MODALITY_UNSPECIFIED", 0), ("IMAGE", 1), ("AUDIO", 2), ("VIDEO", 3)])
_cf.build(_pool)

# --------------------------------------------------------------------------
# sglang_scheduler.proto (reference crates/grpc_client/proto/sglang_scheduler.proto)
# --------------------------------------------------------------------------
_sf = _File("sglang_scheduler.proto", S,
            deps=("google/protobuf/timestamp.proto", "google/protobuf/struct.proto",
                  "common.proto"))

_m = _sf.message("SamplingParams")
_m.field("temperature", 1, T_F32).field("top_p", 2, T_F32).field("top_k", 3, T_I32) \
 .field("min_p", 4, T_F32).field("frequency_penalty", 5, T_F32) \
 .field("presence_penalty", 6, T_F32).field("repetition_penalty", 7, T_F32) \
 .field("max_new_tokens", 8, T_U32, optional=True) \
 .field("stop", 9, T_STR, repeated=True) \
 .field("stop_token_ids", 10, T_U32, repeated=True) \
 .field("skip_special_tokens", 11, T_BOOL) \
 .field("spaces_between_special_tokens", 12, T_BOOL)
_o = _m.oneof("constraint")
_m.field("regex", 13, T_STR, oneof=_o).field("json_schema", 14, T_STR, oneof=_o) \
 .field("ebnf_grammar", 15, T_STR, oneof=_o).field("structural_tag", 16, T_STR, oneof=_o)
_m.field("n", 17, T_U32).field("min_new_tokens", 18, T_U32).field("ignore_eos", 19, T_BOOL) \
 .field("no_stop_trim", 20, T_BOOL).field("stream_interval", 21, T_I32, optional=True)
_m.map_field("logit_bias", 22, T_STR, T_F32)
_m.field("custom_params", 23, T_MSG, type_name=".google.protobuf.Struct")

_sf.message("DisaggregatedParams") \
    .field("bootstrap_host", 1, T_STR).field("bootstrap_port", 2, T_I32) \
    .field("bootstrap_room", 3, T_I32)

_sf.message("TokenizedInput") \
    .field("original_text", 1, T_STR).field("input_ids", 2, T_U32, repeated=True)

_sf.message("TensorData") \
    .field("data", 1, T_BYTES).field("shape", 2, T_U32, repeated=True) \
    .field("dtype", 3, T_STR)

_sf.message("PlaceholderRange").field("offset", 1, T_U32).field("length", 2, T_U32)

_m = _sf.message("MultimodalInputs")
_m.field("image_urls", 1, T_STR, repeated=True) \
 .field("video_urls", 2, T_STR, repeated=True) \
 .field("audio_urls", 3, T_STR, repeated=True) \
 .field("pixel_values", 4, T_MSG, type_name=f".{S}.TensorData") \
 .field("image_data", 5, T_BYTES, repeated=True) \
 .field("video_data", 6, T_BYTES, repeated=True) \
 .field("audio_data", 7, T_BYTES, repeated=True) \
 .field("modalities", 8, T_STR, repeated=True)
_m.map_field("model_specific_tensors", 9, T_STR, T_MSG, val_type_name=f".{S}.TensorData")
_m.field("im_token_id", 10, T_U32, optional=True) \
 .field("mm_placeholders", 11, T_MSG, repeated=True, type_name=f".{S}.PlaceholderRange")

_sf.message("GenerateRequest") \
    .field("request_id", 1, T_STR) \
    .field("tokenized", 2, T_MSG, type_name=f".{S}.TokenizedInput") \
    .field("mm_inputs", 3, T_MSG, type_name=f".{S}.MultimodalInputs") \
    .field("sampling_params", 4, T_MSG, type_name=f".{S}.SamplingParams") \
    .field("return_logprob", 5, T_BOOL) \
    .field("logprob_start_len", 6, T_I32) \
    .field("top_logprobs_num", 7, T_I32) \
    .field("token_ids_logprob", 8, T_U32, repeated=True) \
    .field("return_hidden_states", 9, T_BOOL) \
    .field("disaggregated_params", 10, T_MSG, type_name=f".{S}.DisaggregatedParams") \
    .field("custom_logit_processor", 11, T_STR) \
    .field("timestamp", 12, T_MSG, type_name=".google.protobuf.Timestamp") \
    .field("log_metrics", 13, T_BOOL) \
    .field("input_embeds", 14, T_F32, repeated=True) \
    .field("lora_id", 15, T_STR) \
    .field("data_parallel_rank", 16, T_I32) \
    .field("stream", 17, T_BOOL) \
    .field("require_reasoning", 18, T_BOOL)

_m = _sf.message("GenerateResponse")
_m.field("request_id", 1, T_STR)
_o = _m.oneof("response")
_m.field("chunk", 2, T_MSG, type_name=f".{S}.GenerateStreamChunk", oneof=_o)
_m.field("complete", 3, T_MSG, type_name=f".{S}.GenerateComplete", oneof=_o)

_sf.message("GenerateStreamChunk") \
    .field("token_ids", 1, T_U32, repeated=True) \
    .field("prompt_tokens", 2, T_U32).field("completion_tokens", 3, T_U32) \
    .field("cached_tokens", 4, T_U32) \
    .field("output_logprobs", 5, T_MSG, type_name=f".{S}.OutputLogProbs") \
    .field("hidden_states", 6, T_F32, repeated=True) \
    .field("input_logprobs", 7, T_MSG, type_name=f".{S}.InputLogProbs") \
    .field("index", 8, T_U32).field("reasoning_tokens", 9, T_U32)

_m = _sf.message("GenerateComplete")
_m.field("output_ids", 1, T_U32, repeated=True).field("finish_reason", 2, T_STR) \
 .field("prompt_tokens", 3, T_U32).field("completion_tokens", 4, T_U32) \
 .field("cached_tokens", 5, T_U32) \
 .field("output_logprobs", 6, T_MSG, type_name=f".{S}.OutputLogProbs") \
 .field("all_hidden_states", 7, T_MSG, repeated=True, type_name=f".{S}.HiddenStates")
_o = _m.oneof("matched_stop")
_m.field("matched_token_id", 8, T_U32, oneof=_o) \
 .field("matched_stop_str", 9, T_STR, oneof=_o)
_m.field("input_logprobs", 10, T_MSG, type_name=f".{S}.InputLogProbs") \
 .field("index", 11, T_U32).field("reasoning_tokens", 12, T_U32)

_sf.message("OutputLogProbs") \
    .field("token_logprobs", 1, T_F32, repeated=True) \
    .field("token_ids", 2, T_U32, repeated=True) \
    .field("top_logprobs", 3, T_MSG, repeated=True, type_name=f".{S}.TopLogProbs")
_sf.message("InputLogProbs") \
    .field("token_logprobs", 1, T_MSG, repeated=True, type_name=f".{S}.InputTokenLogProb") \
    .field("token_ids", 2, T_U32, repeated=True) \
    .field("top_logprobs", 3, T_MSG, repeated=True, type_name=f".{S}.TopLogProbs")
_sf.message("InputTokenLogProb").field("value", 1, T_F32, optional=True)
_sf.message("TopLogProbs") \
    .field("values", 1, T_F32, repeated=True).field("token_ids", 2, T_U32, repeated=True)
_sf.message("HiddenStates") \
    .field("values", 1, T_F32, repeated=True).field("layer", 2, T_I32) \
    .field("position", 3, T_I32)

_sf.message("EmbedRequest") \
    .field("request_id", 1, T_STR) \
    .field("tokenized", 2, T_MSG, type_name=f".{S}.TokenizedInput") \
    .field("mm_inputs", 4, T_MSG, type_name=f".{S}.MultimodalInputs") \
    .field("sampling_params", 5, T_MSG, type_name=f".{S}.SamplingParams") \
    .field("token_type_ids", 7, T_I32, repeated=True) \
    .field("data_parallel_rank", 8, T_I32) \
    .field("is_cross_encoder", 9, T_BOOL) \
    .field("texts", 10, T_STR, repeated=True)
_sf.message("EmbedResponse") \
    .field("embedding_dim", 4, T_U32) \
    .field("embedding", 5, T_F32, repeated=True) \
    .field("prompt_tokens", 6, T_U32)

_sf.message("HealthCheckRequest")
_sf.message("HealthCheckResponse").field("healthy", 1, T_BOOL).field("message", 2, T_STR)
_sf.message("AbortRequest").field("request_id", 1, T_STR).field("reason", 2, T_STR)
_sf.message("AbortResponse").field("success", 1, T_BOOL).field("message", 2, T_STR)

_sf.message("LoadLoRAAdapterRequest") \
    .field("lora_name", 1, T_STR).field("lora_path", 2, T_STR) \
    .field("pinned", 3, T_BOOL).field("lora_id", 4, T_STR)
_sf.message("LoadLoRAAdapterResponse") \
    .field("success", 1, T_BOOL).field("message", 2, T_STR) \
    .field("loaded_lora_ids", 3, T_STR, repeated=True)
_sf.message("UnloadLoRAAdapterRequest") \
    .field("lora_name", 1, T_STR).field("lora_id", 2, T_STR)
_sf.message("UnloadLoRAAdapterResponse") \
    .field("success", 1, T_BOOL).field("message", 2, T_STR) \
    .field("loaded_lora_ids", 3, T_STR, repeated=True)
_sf.message("ListLoadedLoRAAdaptersRequest")
_sf.message("LoadedLoRAAdapter") \
    .field("lora_id", 1, T_STR).field("lora_name", 2, T_STR) \
    .field("lora_path", 3, T_STR).field("pinned", 4, T_BOOL)
_sf.message("ListLoadedLoRAAdaptersResponse") \
    .field("loaded_adapters", 1, T_MSG, repeated=True, type_name=f".{S}.LoadedLoRAAdapter")

_sf.message("GetModelInfoRequest")
_sf.message("GetModelInfoResponse") \
    .field("model_path", 1, T_STR).field("tokenizer_path", 2, T_STR) \
    .field("is_generation", 3, T_BOOL).field("preferred_sampling_params", 4, T_STR) \
    .field("weight_version", 5, T_STR).field("served_model_name", 6, T_STR) \
    .field("max_context_length", 7, T_I32).field("vocab_size", 8, T_I32) \
    .field("supports_vision", 9, T_BOOL).field("model_type", 10, T_STR) \
    .field("eos_token_ids", 11, T_I32, repeated=True).field("pad_token_id", 12, T_I32) \
    .field("bos_token_id", 13, T_I32).field("max_req_input_len", 14, T_I32) \
    .field("architectures", 15, T_STR, repeated=True) \
    .field("id2label_json", 16, T_STR).field("num_labels", 17, T_I32) \
    .field("default_sampling_params_json", 20, T_STR)

_sf.message("GetServerInfoRequest")
_sf.message("GetServerInfoResponse") \
    .field("server_args", 1, T_MSG, type_name=".google.protobuf.Struct") \
    .field("scheduler_info", 2, T_MSG, type_name=".google.protobuf.Struct") \
    .field("active_requests", 3, T_I32).field("is_paused", 4, T_BOOL) \
    .field("last_receive_timestamp", 5, T_F64).field("uptime_seconds", 6, T_F64) \
    .field("sglang_version", 7, T_STR).field("server_type", 8, T_STR) \
    .field("start_time", 9, T_MSG, type_name=".google.protobuf.Timestamp") \
    .field("max_total_num_tokens", 10, T_I32)

_sf.message("GetLoadsRequest") \
    .field("dp_rank", 1, T_I32, optional=True) \
    .field("include", 2, T_STR, repeated=True)
_sf.message("GetLoadsResponse") \
    .field("timestamp", 1, T_STR).field("version", 2, T_STR) \
    .field("dp_rank_count", 3, T_I32) \
    .field("loads", 4, T_MSG, repeated=True, type_name=f".{S}.SchedulerLoad") \
    .field("aggregate", 5, T_MSG, type_name=f".{S}.AggregateMetrics")
_sf.message("SchedulerLoad") \
    .field("dp_rank", 1, T_I32).field("num_running_reqs", 2, T_I32) \
    .field("num_waiting_reqs", 3, T_I32).field("num_total_reqs", 4, T_I32) \
    .field("num_used_tokens", 5, T_I32).field("max_total_num_tokens", 6, T_I32) \
    .field("token_usage", 7, T_F64).field("gen_throughput", 8, T_F64) \
    .field("cache_hit_rate", 9, T_F64).field("utilization", 10, T_F64) \
    .field("max_running_requests", 11, T_I32) \
    .field("memory", 12, T_MSG, type_name=f".{S}.MemoryMetrics", optional=True) \
    .field("speculative", 13, T_MSG, type_name=f".{S}.SpeculativeMetrics", optional=True) \
    .field("lora", 14, T_MSG, type_name=f".{S}.LoRAMetrics", optional=True) \
    .field("disaggregation", 15, T_MSG, type_name=f".{S}.DisaggregationMetrics", optional=True) \
    .field("queues", 16, T_MSG, type_name=f".{S}.QueueMetrics", optional=True) \
    .field("num_waiting_uncached_tokens", 17, T_I32)
_sf.message("MemoryMetrics") \
    .field("weight_gb", 1, T_F64).field("kv_cache_gb", 2, T_F64) \
    .field("graph_gb", 3, T_F64).field("token_capacity", 4, T_I32)
_sf.message("SpeculativeMetrics") \
    .field("accept_length", 1, T_F64).field("accept_rate", 2, T_F64)
_sf.message("LoRAMetrics") \
    .field("slots_used", 1, T_I32).field("slots_total", 2, T_I32) \
    .field("utilization", 3, T_F64)
_sf.message("DisaggregationMetrics") \
    .field("mode", 1, T_STR).field("prefill_prealloc_queue_reqs", 2, T_I32) \
    .field("prefill_inflight_queue_reqs", 3, T_I32) \
    .field("decode_prealloc_queue_reqs", 4, T_I32) \
    .field("decode_transfer_queue_reqs", 5, T_I32) \
    .field("decode_retracted_queue_reqs", 6, T_I32) \
    .field("kv_transfer_speed_gb_s", 7, T_F64) \
    .field("kv_transfer_latency_ms", 8, T_F64)
_sf.message("QueueMetrics") \
    .field("waiting", 1, T_I32).field("grammar", 2, T_I32) \
    .field("paused", 3, T_I32).field("retracted", 4, T_I32)
_sf.message("AggregateMetrics") \
    .field("total_running_reqs", 1, T_I32).field("total_waiting_reqs", 2, T_I32) \
    .field("total_reqs", 3, T_I32).field("avg_token_usage", 4, T_F64) \
    .field("avg_throughput", 5, T_F64).field("avg_utilization", 6, T_F64)
_sf.build(_pool)


def _cls(full_name: str):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(full_name))


# public message classes -----------------------------------------------------
_COMMON_NAMES = [
    "GetTokenizerRequest", "GetTokenizerChunk", "SubscribeKvEventsRequest",
    "KvEventBatch", "KvCacheEvent", "KvBlocksStored", "KvBlock", "KvBlocksRemoved",
    "KvCacheCleared", "FlushCacheRequest", "FlushCacheResponse", "StartProfileRequest",
    "StopProfileRequest", "ProfileResponse", "ShmHandle", "RemoteTensorHandle",
]
_SCHED_NAMES = [
    "SamplingParams", "DisaggregatedParams", "TokenizedInput", "TensorData",
    "PlaceholderRange", "MultimodalInputs", "GenerateRequest", "GenerateResponse",
    "GenerateStreamChunk", "GenerateComplete", "OutputLogProbs", "InputLogProbs",
    "InputTokenLogProb", "TopLogProbs", "HiddenStates", "EmbedRequest", "EmbedResponse",
    "HealthCheckRequest", "HealthCheckResponse", "AbortRequest", "AbortResponse",
    "LoadLoRAAdapterRequest", "LoadLoRAAdapterResponse", "UnloadLoRAAdapterRequest",
    "UnloadLoRAAdapterResponse", "ListLoadedLoRAAdaptersRequest", "LoadedLoRAAdapter",
    "ListLoadedLoRAAdaptersResponse", "GetModelInfoRequest", "GetModelInfoResponse",
    "GetServerInfoRequest", "GetServerInfoResponse", "GetLoadsRequest",
    "GetLoadsResponse", "SchedulerLoad", "MemoryMetrics", "SpeculativeMetrics",
    "LoRAMetrics", "DisaggregationMetrics", "QueueMetrics", "AggregateMetrics",
]

MESSAGES: Dict[str, type] = {}
for _n in _COMMON_NAMES:
    MESSAGES[_n] = _cls(f"{C}.{_n}")
for _n in _SCHED_NAMES:
    MESSAGES[_n] = _cls(f"{S}.{_n}")

globals().update(MESSAGES)

SERVICE_NAME = "sglang.grpc.scheduler.SglangScheduler"

# method -> (request class, response class, server_streaming)
METHODS = {
    "Generate": (MESSAGES["GenerateRequest"], MESSAGES["GenerateResponse"], True),
    "Embed": (MESSAGES["EmbedRequest"], MESSAGES["EmbedResponse"], False),
    "HealthCheck": (MESSAGES["HealthCheckRequest"], MESSAGES["HealthCheckResponse"], False),
    "Abort": (MESSAGES["AbortRequest"], MESSAGES["AbortResponse"], False),
    "GetModelInfo": (MESSAGES["GetModelInfoRequest"], MESSAGES["GetModelInfoResponse"], False),
    "GetServerInfo": (MESSAGES["GetServerInfoRequest"], MESSAGES["GetServerInfoResponse"], False),
    "GetLoads": (MESSAGES["GetLoadsRequest"], MESSAGES["GetLoadsResponse"], False),
    "FlushCache": (MESSAGES["FlushCacheRequest"], MESSAGES["FlushCacheResponse"], False),
    "StartProfile": (MESSAGES["StartProfileRequest"], MESSAGES["ProfileResponse"], False),
    "StopProfile": (MESSAGES["StopProfileRequest"], MESSAGES["ProfileResponse"], False),
    "GetTokenizer": (MESSAGES["GetTokenizerRequest"], MESSAGES["GetTokenizerChunk"], True),
    "SubscribeKvEvents": (MESSAGES["SubscribeKvEventsRequest"], MESSAGES["KvEventBatch"], True),
    "LoadLoRAAdapter": (MESSAGES["LoadLoRAAdapterRequest"], MESSAGES["LoadLoRAAdapterResponse"], False),
    "UnloadLoRAAdapter": (MESSAGES["UnloadLoRAAdapterRequest"], MESSAGES["UnloadLoRAAdapterResponse"], False),
    "ListLoadedLoRAAdapters": (MESSAGES["ListLoadedLoRAAdaptersRequest"],
                               MESSAGES["ListLoadedLoRAAdaptersResponse"], False),
}


def method_path(name: str) -> str:
    return f"/{SERVICE_NAME}/{name}"
