"""UDS tokenizer client + bounded tokenization worker pool.

Capability parity with the reference pkg/tokenization (pool.go,
uds_tokenizer.go): gRPC-over-UDS client with keepalive and 100 MB message
caps, model -> tokenizer-path mapping, and a worker pool whose blocking
``tokenize`` retries transient failures before giving up.
"""
from __future__ import annotations

import concurrent.futures
import logging
from typing import Dict, List, Optional, Sequence, Tuple

import grpc

from . import proto

log = logging.getLogger(__name__)

MAX_MSG = 100 * 1024 * 1024
DEFAULT_TIMEOUT_S = 5.0


class UdsTokenizerClient:
    def __init__(self, uds_path: str,
                 model_paths: Optional[Dict[str, str]] = None,
                 timeout_s: float = DEFAULT_TIMEOUT_S):
        self._channel = grpc.insecure_channel(
            f"unix://{uds_path}",
            options=[
                ("grpc.max_receive_message_length", MAX_MSG),
                ("grpc.max_send_message_length", MAX_MSG),
                ("grpc.keepalive_time_ms", 30000),
            ],
        )
        self._timeout = timeout_s
        self._paths = dict(model_paths or {})
        svc = "tokenizerpb.TokenizationService"
        g = proto.get
        self._init = self._channel.unary_unary(
            f"/{svc}/InitializeTokenizer",
            request_serializer=g("tokenizerpb.InitializeTokenizerRequest").SerializeToString,
            response_deserializer=g("tokenizerpb.InitializeTokenizerResponse").FromString,
        )
        self._tok = self._channel.unary_unary(
            f"/{svc}/Tokenize",
            request_serializer=g("tokenizerpb.TokenizeRequest").SerializeToString,
            response_deserializer=g("tokenizerpb.TokenizeResponse").FromString,
        )
        self._render = self._channel.unary_unary(
            f"/{svc}/RenderChatTemplate",
            request_serializer=g("tokenizerpb.RenderChatRequest").SerializeToString,
            response_deserializer=g("tokenizerpb.RenderChatResponse").FromString,
        )
        self._render_cc = self._channel.unary_unary(
            f"/{svc}/RenderChatCompletion",
            request_serializer=g(
                "tokenizerpb.RenderChatCompletionRequest").SerializeToString,
            response_deserializer=g(
                "tokenizerpb.RenderChatCompletionResponse").FromString,
        )
        self._render_c = self._channel.unary_unary(
            f"/{svc}/RenderCompletion",
            request_serializer=g(
                "tokenizerpb.RenderCompletionRequest").SerializeToString,
            response_deserializer=g(
                "tokenizerpb.RenderCompletionResponse").FromString,
        )

    def initialize(self, model_name: str, tokenizer_path: str = "") -> bool:
        Req = proto.get("tokenizerpb.InitializeTokenizerRequest")
        path = tokenizer_path or self._paths.get(model_name, "")
        resp = self._init(Req(model_name=model_name, tokenizer_path=path),
                          timeout=30.0)
        if not resp.success:
            raise RuntimeError(f"tokenizer init failed: {resp.error}")
        return True

    def tokenize(self, model_name: str, text: str,
                 add_special_tokens: bool = True) -> List[int]:
        Req = proto.get("tokenizerpb.TokenizeRequest")
        resp = self._tok(
            Req(model_name=model_name, text=text,
                add_special_tokens=add_special_tokens),
            timeout=self._timeout,
        )
        if resp.error:
            raise RuntimeError(resp.error)
        return list(resp.token_ids)

    def render_chat(self, model_name: str,
                    messages: Sequence[Tuple[str, str]],
                    add_generation_prompt: bool = True) -> Tuple[List[int], str]:
        Req = proto.get("tokenizerpb.RenderChatRequest")
        Msg = proto.get("tokenizerpb.ChatMessage")
        resp = self._render(
            Req(model_name=model_name,
                messages=[Msg(role=r, content=c) for r, c in messages],
                add_generation_prompt=add_generation_prompt),
            timeout=max(self._timeout, 30.0),  # multimodal renders take longer
        )
        if resp.error:
            raise RuntimeError(resp.error)
        return list(resp.token_ids), resp.rendered

    def render_chat_completion(self, model_name: str, messages,
                               add_generation_prompt: Optional[bool] = None,
                               continue_final_message: bool = False,
                               chat_template: str = "",
                               chat_template_kwargs: str = ""):
        """OpenAI chat-completion render. messages: dicts with role +
        content (str) or content_parts (list of {type, text|image_url}).
        Returns (request_id, token_ids, mm_hashes, placeholder_ranges)."""
        Req = proto.get("tokenizerpb.RenderChatCompletionRequest")
        Msg = proto.get("tokenizerpb.ChatMessage")
        Part = proto.get("tokenizerpb.ContentPart")
        msgs = []
        for m in messages:
            pm = Msg(role=m.get("role", "user"), content=m.get("content", ""))
            for part in m.get("content_parts", []) or []:
                p = Part(type=part.get("type", "text"),
                         text=part.get("text", "") or "")
                if part.get("image_url"):
                    p.image_url.url = part["image_url"].get("url", "")
                pm.content_parts.append(p)
            msgs.append(pm)
        req = Req(model_name=model_name, messages=msgs,
                  continue_final_message=continue_final_message,
                  chat_template=chat_template,
                  chat_template_kwargs=chat_template_kwargs)
        if add_generation_prompt is not None:
            req.add_generation_prompt = add_generation_prompt
        resp = self._render_cc(req, timeout=max(self._timeout, 30.0))
        if resp.error:
            raise RuntimeError(resp.error)
        hashes = list(resp.features.mm_hashes)
        ranges = [(r.offset, r.length)
                  for r in resp.features.placeholder_ranges]
        return resp.request_id, list(resp.token_ids), hashes, ranges

    def render_completion(self, model_name: str, prompt: str):
        """OpenAI completion render: (request_id, token_ids)."""
        Req = proto.get("tokenizerpb.RenderCompletionRequest")
        resp = self._render_c(Req(model_name=model_name, prompt=prompt),
                              timeout=self._timeout)
        if resp.error:
            raise RuntimeError(resp.error)
        return resp.request_id, list(resp.token_ids)

    def close(self):
        self._channel.close()


class TokenizationPool:
    """Bounded worker pool over the UDS client; blocking tokenize with
    bounded retries (reference pool.go:100-130 drops after 3 attempts)."""

    def __init__(self, client: UdsTokenizerClient, workers: int = 5,
                 retries: int = 3):
        self._client = client
        self._pool = concurrent.futures.ThreadPoolExecutor(max_workers=workers)
        self._retries = retries

    def tokenize(self, model_name: str, text: str, timeout: float = 30.0) -> List[int]:
        fut = self._pool.submit(self._tokenize_with_retry, model_name, text)
        return fut.result(timeout=timeout)

    def tokenize_async(self, model_name: str, text: str):
        return self._pool.submit(self._tokenize_with_retry, model_name, text)

    def _tokenize_with_retry(self, model_name: str, text: str) -> List[int]:
        last = None
        for attempt in range(self._retries):
            try:
                return self._client.tokenize(model_name, text)
            except Exception as e:  # transient UDS failures
                last = e
                log.warning("tokenize attempt %d failed: %s", attempt + 1, e)
        raise RuntimeError(f"tokenization failed after {self._retries} attempts: {last}")

    def shutdown(self):
        self._pool.shutdown(wait=False)
