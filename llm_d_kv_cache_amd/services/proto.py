"""Runtime protobuf descriptors for api/indexer.proto + api/tokenizer.proto.

The build image ships the protobuf runtime but no protoc, so the message
classes are constructed from FileDescriptorProto built in code. The .proto
files under api/ are the authoritative schema; this module mirrors them
field-for-field (numbers included) so the wire format matches any
protoc-generated peer.
"""
from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_F = descriptor_pb2.FieldDescriptorProto

_POOL = descriptor_pool.Default()
_CLASSES = {}


def _msg(fdp, name, fields):
    m = fdp.message_type.add()
    m.name = name
    for fname, num, ftype, label, type_name in fields:
        f = m.field.add()
        f.name = fname
        f.number = num
        f.type = ftype
        if label == "POPT":  # proto3 `optional` (explicit presence)
            f.label = _F.LABEL_OPTIONAL
            f.proto3_optional = True
            od = m.oneof_decl.add()
            od.name = "_" + fname
            f.oneof_index = len(m.oneof_decl) - 1
        else:
            f.label = label
        if type_name:
            f.type_name = type_name
    return m


def _build():
    if _CLASSES:
        return
    OPT, REP = _F.LABEL_OPTIONAL, _F.LABEL_REPEATED
    S, D, U32, U64, B = (_F.TYPE_STRING, _F.TYPE_DOUBLE, _F.TYPE_UINT32,
                         _F.TYPE_UINT64, _F.TYPE_BOOL)
    M = _F.TYPE_MESSAGE

    idx = descriptor_pb2.FileDescriptorProto()
    idx.name = "kvcache_amd/indexer.proto"
    idx.package = "indexerpb"
    idx.syntax = "proto3"
    _msg(idx, "ScoreRequest", [
        ("prompt", 1, S, OPT, None),
        ("model_name", 2, S, OPT, None),
        ("pod_identifiers", 3, S, REP, None),
    ])
    _msg(idx, "ScoreTokensRequest", [
        ("tokens", 1, U32, REP, None),
        ("model_name", 2, S, OPT, None),
        ("pod_identifiers", 3, S, REP, None),
    ])
    _msg(idx, "PodScore", [
        ("pod_identifier", 1, S, OPT, None),
        ("score", 2, D, OPT, None),
    ])
    _msg(idx, "ScoreResponse", [
        ("scores", 1, M, REP, ".indexerpb.PodScore"),
        ("total_blocks", 2, U64, OPT, None),
        ("hit_blocks", 3, U64, OPT, None),
    ])

    tok = descriptor_pb2.FileDescriptorProto()
    tok.name = "kvcache_amd/tokenizer.proto"
    tok.package = "tokenizerpb"
    tok.syntax = "proto3"
    _msg(tok, "InitializeTokenizerRequest", [
        ("model_name", 1, S, OPT, None),
        ("tokenizer_path", 2, S, OPT, None),
    ])
    _msg(tok, "InitializeTokenizerResponse", [
        ("success", 1, B, OPT, None),
        ("error", 2, S, OPT, None),
    ])
    _msg(tok, "TokenizeRequest", [
        ("model_name", 1, S, OPT, None),
        ("text", 2, S, OPT, None),
        ("add_special_tokens", 3, B, OPT, None),
    ])
    _msg(tok, "OffsetMapping", [
        ("start", 1, U32, OPT, None),
        ("end", 2, U32, OPT, None),
    ])
    _msg(tok, "TokenizeResponse", [
        ("token_ids", 1, U32, REP, None),
        ("offsets", 2, M, REP, ".tokenizerpb.OffsetMapping"),
        ("error", 3, S, OPT, None),
    ])
    _msg(tok, "ChatMessage", [
        ("role", 1, S, OPT, None),
        ("content", 2, S, OPT, None),
        ("content_parts", 3, M, REP, ".tokenizerpb.ContentPart"),
    ])
    _msg(tok, "PlaceholderRange", [
        ("offset", 1, U64, OPT, None),
        ("length", 2, U64, OPT, None),
    ])
    _msg(tok, "MultiModalFeatures", [
        ("mm_hashes", 1, S, REP, None),
        ("placeholder_ranges", 2, M, REP, ".tokenizerpb.PlaceholderRange"),
    ])
    _msg(tok, "RenderChatRequest", [
        ("model_name", 1, S, OPT, None),
        ("messages", 2, M, REP, ".tokenizerpb.ChatMessage"),
        ("add_generation_prompt", 3, B, OPT, None),
        ("mm_item_hashes", 4, S, REP, None),
    ])
    _msg(tok, "RenderChatResponse", [
        ("token_ids", 1, U32, REP, None),
        ("rendered", 2, S, OPT, None),
        ("mm_features", 3, M, OPT, ".tokenizerpb.MultiModalFeatures"),
        ("error", 4, S, OPT, None),
    ])
    _msg(tok, "ImageUrl", [
        ("url", 1, S, OPT, None),
    ])
    _msg(tok, "ContentPart", [
        ("type", 1, S, OPT, None),
        ("text", 2, S, OPT, None),
        ("image_url", 3, M, OPT, ".tokenizerpb.ImageUrl"),
    ])
    _msg(tok, "RenderChatCompletionRequest", [
        ("model_name", 1, S, OPT, None),
        ("messages", 2, M, REP, ".tokenizerpb.ChatMessage"),
        ("tools_json", 3, S, OPT, None),
        ("chat_template", 4, S, OPT, None),
        ("add_generation_prompt", 5, B, "POPT", None),
        ("continue_final_message", 6, B, OPT, None),
        ("chat_template_kwargs", 7, S, OPT, None),
    ])
    _msg(tok, "RenderChatCompletionResponse", [
        ("request_id", 1, S, OPT, None),
        ("token_ids", 2, U32, REP, None),
        ("features", 3, M, OPT, ".tokenizerpb.MultiModalFeatures"),
        ("error", 4, S, OPT, None),
    ])
    _msg(tok, "RenderCompletionRequest", [
        ("model_name", 1, S, OPT, None),
        ("prompt", 2, S, OPT, None),
    ])
    _msg(tok, "RenderCompletionResponse", [
        ("request_id", 1, S, OPT, None),
        ("token_ids", 2, U32, REP, None),
        ("error", 3, S, OPT, None),
    ])

    for fdp in (idx, tok):
        try:
            _POOL.Add(fdp)
        except Exception:
            pass  # already registered (re-import)
        for m in fdp.message_type:
            full = f"{fdp.package}.{m.name}"
            _CLASSES[full] = message_factory.GetMessageClass(
                _POOL.FindMessageTypeByName(full)
            )


def get(name: str):
    """Message class by full name, e.g. 'indexerpb.ScoreRequest'."""
    _build()
    return _CLASSES[name]
