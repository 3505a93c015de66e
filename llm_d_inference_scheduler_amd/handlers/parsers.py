"""Request/response parsers (L2)
(parity: pkg/epp/framework/plugins/requesthandling/parsers/* and the parser
mux pkg/epp/handlers/parsers.go:28; unified body type
requesthandling/types.go:65).

* openai-parser: /v1/completions, /v1/chat/completions, /v1/responses,
  /v1/conversations, /v1/embeddings JSON -> LLMRequest; SSE + JSON usage
  parse on the response side (parsers/openai/openai.go:34-139).
* passthrough-parser: raw bytes, Skip -> fallback random-endpoint routing.
* vertexai-parser: Google Vertex AI payload shape.
* vllm-grpc-parser: direct protobuf wire-format decode of the VllmEngine
  service's Generate/Embed requests + usage responses (no codegen);
  non-routable methods Skip to random-endpoint fallback.
"""
import json
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..scheduling.types import LLMRequest, MultiModalItem
from ..plugins.registry import register_plugin
from ..plugins.interface import Plugin

FAIRNESS_ID_HEADER = "x-gateway-inference-fairness-id"
OBJECTIVE_HEADER = "x-gateway-inference-objective"
MODEL_REWRITE_HEADER = "x-gateway-model-name-rewrite"


@dataclass
class Usage:
    prompt_tokens: int = 0
    completion_tokens: int = 0
    cached_tokens: int = 0
    ttft_ms: Optional[float] = None
    tpot_ms: Optional[float] = None
    e2e_ms: Optional[float] = None

    def to_openai(self) -> Dict[str, Any]:
        return {
            "prompt_tokens": self.prompt_tokens,
            "completion_tokens": self.completion_tokens,
            "total_tokens": self.prompt_tokens + self.completion_tokens,
            "prompt_tokens_details": {"cached_tokens": self.cached_tokens},
        }


@dataclass
class ParseResult:
    request: Optional[LLMRequest] = None
    skip: bool = False          # Skip -> random-endpoint fallback routing
    error: Optional[str] = None


class Parser(Plugin):
    def parse_request(self, body: bytes, headers: Dict[str, str],
                      path: str = "") -> ParseResult:
        raise NotImplementedError

    def parse_response_usage(self, body: bytes,
                             streaming: bool) -> Optional[Usage]:
        return None


def _headers_into(req: LLMRequest, headers: Dict[str, str]) -> None:
    req.headers.update(headers)
    req.fairness_id = headers.get(FAIRNESS_ID_HEADER, "")
    req.objective_name = headers.get(OBJECTIVE_HEADER, "")
    # explicit per-request rewrite wins over InferenceModelRewrite rules
    # (request.go:56 ModelNameRewriteKey -> TargetModelName)
    forced = headers.get(MODEL_REWRITE_HEADER, "")
    if forced:
        req.target_model = forced


def _extract_mm_items(messages: List[Dict[str, Any]]) -> List[MultiModalItem]:
    items: List[MultiModalItem] = []
    for msg in messages:
        content = msg.get("content")
        if not isinstance(content, list):
            continue
        for part in content:
            kind = part.get("type", "")
            if kind == "image_url":
                url = part.get("image_url", {})
                items.append(MultiModalItem("image_url",
                                            url.get("url", "") if isinstance(url, dict) else str(url)))
            elif kind == "video_url":
                items.append(MultiModalItem("video_url",
                                            part.get("video_url", {}).get("url", "")))
            elif kind == "input_audio":
                items.append(MultiModalItem("input_audio",
                                            part.get("input_audio", {}).get("data", "")[:64]))
    return items


@register_plugin("openai-parser")
class OpenAIParser(Parser):
    def parse_request(self, body: bytes, headers: Dict[str, str],
                      path: str = "") -> ParseResult:
        try:
            data = json.loads(body)
        except (ValueError, TypeError) as e:
            return ParseResult(error=f"invalid JSON: {e}")
        if not isinstance(data, dict) or "model" not in data:
            return ParseResult(error="missing model")
        req = LLMRequest(
            request_id=headers.get("x-request-id", uuid.uuid4().hex),
            model=str(data["model"]),
            raw_body=data,
        )
        if "/embeddings" in path or "input" in data and "messages" not in data \
                and "prompt" not in data and path.endswith("embeddings"):
            req.is_embedding = True
        if "prompt" in data:
            p = data["prompt"]
            req.prompt = p if isinstance(p, str) else " ".join(map(str, p))
        if "messages" in data and isinstance(data["messages"], list):
            req.messages = data["messages"]
            req.mm_items = _extract_mm_items(req.messages)
        if "input" in data and not req.prompt and not req.messages:
            inp = data["input"]
            req.prompt = inp if isinstance(inp, str) else " ".join(map(str, inp))
            req.is_embedding = "/embeddings" in path
        req.max_tokens = int(data.get("max_tokens",
                                      data.get("max_completion_tokens", 16)))
        req.temperature = float(data.get("temperature", 0.0))
        if isinstance(data.get("stop_token_ids"), list):
            req.stop_token_ids = [int(t) for t in data["stop_token_ids"]]
        req.streaming = bool(data.get("stream", False))
        req.session_id = str(data.get("user", "") or "")
        _headers_into(req, headers)
        return ParseResult(request=req)

    def parse_response_usage(self, body: bytes,
                             streaming: bool) -> Optional[Usage]:
        """JSON body or SSE stream -> usage (response.go:45-63)."""
        try:
            if streaming:
                usage = None
                for line in body.split(b"\n"):
                    line = line.strip()
                    if not line.startswith(b"data:"):
                        continue
                    payload = line[5:].strip()
                    if payload == b"[DONE]":
                        break
                    d = json.loads(payload)
                    if d.get("usage"):
                        usage = d["usage"]
                d = {"usage": usage} if usage else {}
            else:
                d = json.loads(body)
            u = d.get("usage") or {}
            if not u:
                return None
            details = u.get("prompt_tokens_details") or {}
            return Usage(prompt_tokens=int(u.get("prompt_tokens", 0)),
                         completion_tokens=int(u.get("completion_tokens", 0)),
                         cached_tokens=int(details.get("cached_tokens", 0)))
        except (ValueError, TypeError):
            return None


@register_plugin("passthrough-parser")
class PassthroughParser(Parser):
    """Raw bytes, skip parsing -> random-endpoint fallback (parsers/passthrough)."""

    def parse_request(self, body, headers, path=""):
        return ParseResult(skip=True)


@register_plugin("vertexai-parser")
class VertexAIParser(Parser):
    """Google Vertex AI payloads (parsers/vertexai): instances/contents shape."""

    def parse_request(self, body: bytes, headers: Dict[str, str],
                      path: str = "") -> ParseResult:
        try:
            data = json.loads(body)
        except (ValueError, TypeError) as e:
            return ParseResult(error=f"invalid JSON: {e}")
        if not isinstance(data, dict):
            return ParseResult(error="body is not a JSON object")
        model = str(data.get("model", "")) or headers.get("x-vertex-model", "")
        if not model:
            return ParseResult(error="missing model")
        req = LLMRequest(
            request_id=headers.get("x-request-id", uuid.uuid4().hex),
            model=model, raw_body=data)
        contents = data.get("contents") or []
        parts: List[str] = []
        for c in contents if isinstance(contents, list) else []:
            if not isinstance(c, dict):
                continue
            parts_list = c.get("parts", [])
            for part in parts_list if isinstance(parts_list, list) else []:
                if isinstance(part, dict) and "text" in part:
                    parts.append(str(part["text"]))
        req.prompt = "\n".join(parts)
        gen_cfg = data.get("generationConfig") or {}
        req.max_tokens = int(gen_cfg.get("maxOutputTokens", 16))
        req.temperature = float(gen_cfg.get("temperature", 0.0))
        _headers_into(req, headers)
        return ParseResult(request=req)


class ParserMux:
    """Per content-type / path dispatch (handlers/parsers.go:28)."""

    def __init__(self, default: Optional[Parser] = None):
        self.default = default or OpenAIParser()
        self.by_content_type: Dict[str, Parser] = {}

    def register(self, content_type: str, parser: Parser) -> None:
        self.by_content_type[content_type] = parser

    def parse_request(self, body: bytes, headers: Dict[str, str],
                      path: str = "") -> ParseResult:
        ct = headers.get("content-type", "application/json").split(";")[0]
        parser = self.by_content_type.get(ct, self.default)
        return parser.parse_request(body, headers, path)

    def parse_response_usage(self, body: bytes, headers: Dict[str, str],
                             streaming: bool = False) -> Optional[Usage]:
        ct = headers.get("content-type", "application/json").split(";")[0]
        parser = self.by_content_type.get(ct, self.default)
        streaming = streaming or body.lstrip().startswith(b"data:")
        return parser.parse_response_usage(body, streaming)


# ---------------------------------------------------------------------------
# vLLM gRPC parser (parsers/vllmgrpc): the reference parses the vLLM
# `VllmEngine` gRPC service (Generate/Embed/HealthCheck/Abort/GetModelInfo/
# GetServerInfo, vllm_engine.proto:10-27) from generated protobuf stubs.
# Here the protobuf wire format is decoded directly (no codegen): varint /
# 64-bit / length-delimited / 32-bit field walking over the gRPC
# length-prefixed message frame.

def _pb_walk(data: bytes):
    """Yield (field_number, wire_type, value) over a protobuf message."""
    i, n = 0, len(data)
    while i < n:
        tag, i = _pb_varint(data, i)
        field, wt = tag >> 3, tag & 7
        if wt == 0:                      # varint
            val, i = _pb_varint(data, i)
        elif wt == 1:                    # 64-bit
            val, i = data[i:i + 8], i + 8
            if len(val) != 8:
                raise ValueError("truncated fixed64")
        elif wt == 2:                    # length-delimited
            ln, i = _pb_varint(data, i)
            val, i = data[i:i + ln], i + ln
            if len(val) != ln:
                raise ValueError("truncated length-delimited field")
        elif wt == 5:                    # 32-bit
            val, i = data[i:i + 4], i + 4
            if len(val) != 4:
                raise ValueError("truncated fixed32")
        else:
            raise ValueError(f"unsupported wire type {wt}")
        yield field, wt, val


def _pb_varint(data: bytes, i: int):
    shift, out = 0, 0
    while True:
        b = data[i]
        i += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            return out, i
        shift += 7
        if shift > 63:
            raise ValueError("varint overflow")


def _pb_packed_uint32(val: bytes) -> List[int]:
    out, i = [], 0
    while i < len(val):
        v, i = _pb_varint(val, i)
        out.append(v)
    return out


def _strip_grpc_frame(body: bytes) -> bytes:
    """gRPC messages are framed [compressed u8][length u32 BE][payload]."""
    if len(body) >= 5 and body[0] in (0, 1):
        ln = int.from_bytes(body[1:5], "big")
        if len(body) == 5 + ln:
            if body[0] == 1:
                raise ValueError("compressed gRPC frame unsupported")
            return body[5:]
    return body


@register_plugin("vllm-grpc-parser")
class VllmGrpcParser(Parser):
    """vLLM gRPC `VllmEngine` request parser.

    Message schema (field numbers mirror the reference's
    vllm_engine.proto Generate/Embed requests):
      GenerateRequest: 1 model(str) 2 prompt(str) 3 token_ids(packed u32)
                       4 max_tokens(u32) 5 temperature(f32) 6 stream(bool)
                       7 request_id(str)
      EmbedRequest:    1 model(str) 2 prompt(str) 3 token_ids(packed u32)
                       7 request_id(str)
      GenerateResponse usage fields: 1 prompt_tokens 2 completion_tokens
                       3 cached_tokens (varints)
    Non-routable methods (HealthCheck/Abort/GetModelInfo/GetServerInfo)
    return Skip -> random-endpoint fallback, as the reference does for
    parser-skipped bodies (server.go:335-342).
    """

    ROUTABLE = ("/Generate", "/Embed")

    def parse_request(self, body: bytes, headers: Dict[str, str],
                      path: str = "") -> ParseResult:
        method = path.rsplit("/", 1)[-1] if path else "Generate"
        if method not in ("Generate", "Embed"):
            return ParseResult(skip=True)
        try:
            payload = _strip_grpc_frame(body)
            req = LLMRequest(
                request_id=headers.get("x-request-id", uuid.uuid4().hex),
                model="")
            import struct
            for field, wt, val in _pb_walk(payload):
                if field == 1 and wt == 2:
                    req.model = val.decode("utf-8", "replace")
                elif field == 2 and wt == 2:
                    req.prompt = val.decode("utf-8", "replace")
                elif field == 3 and wt == 2:
                    req.prompt_tokens = _pb_packed_uint32(val)
                elif field == 3 and wt == 0:   # unpacked repeated uint32
                    req.prompt_tokens = (req.prompt_tokens or []) + [val]
                elif field == 4 and wt == 0:
                    req.max_tokens = int(val)
                elif field == 5 and wt == 5:
                    req.temperature = struct.unpack("<f", val)[0]
                elif field == 6 and wt == 0:
                    req.streaming = bool(val)
                elif field == 7 and wt == 2:
                    req.request_id = val.decode("utf-8", "replace")
        except (ValueError, IndexError) as e:
            return ParseResult(error=f"invalid protobuf: {e}")
        if not req.model:
            return ParseResult(error="missing model")
        req.is_embedding = method == "Embed"
        _headers_into(req, headers)
        return ParseResult(request=req)

    def parse_response_usage(self, body: bytes,
                             streaming: bool) -> Optional[Usage]:
        try:
            payload = _strip_grpc_frame(body)
            u = Usage()
            for field, wt, val in _pb_walk(payload):
                if wt != 0:
                    continue
                if field == 1:
                    u.prompt_tokens = int(val)
                elif field == 2:
                    u.completion_tokens = int(val)
                elif field == 3:
                    u.cached_tokens = int(val)
            return u
        except (ValueError, IndexError):
            return None
