"""UDS tokenizer sidecar: async gRPC TokenizationService.

Capability parity with the reference services/uds_tokenizer/: loads and
caches HF tokenizers (local paths — offline deployments map model ->
tokenizer directory), serves Tokenize with offset mappings and chat-
template rendering over a unix-domain socket with large message caps.
Implemented with grpc.aio generic handlers (runtime descriptors from
services/proto.py; no protoc in the image). CPU-bound tokenization runs
on a bounded thread pool so the event loop stays responsive.
"""
from __future__ import annotations

import asyncio
import concurrent.futures
import logging
import os
from typing import Dict, Optional

import grpc

from . import proto

log = logging.getLogger(__name__)

MAX_MSG = 100 * 1024 * 1024  # parity with reference uds_tokenizer.go:110-121


class TokenizerManager:
    """Loads and caches tokenizers by model name."""

    def __init__(self, model_paths: Optional[Dict[str, str]] = None,
                 max_workers: int = 8):
        self._paths = dict(model_paths or {})
        self._cache: Dict[str, object] = {}
        self._pool = concurrent.futures.ThreadPoolExecutor(max_workers=max_workers)

    def register(self, model_name: str, path: str) -> None:
        self._paths[model_name] = path
        self._cache.pop(model_name, None)

    def _load(self, model_name: str):
        if model_name in self._cache:
            return self._cache[model_name]
        path = self._paths.get(model_name, model_name)
        from transformers import AutoTokenizer

        tok = AutoTokenizer.from_pretrained(path, local_files_only=True)
        self._cache[model_name] = tok
        return tok

    def tokenize(self, model_name: str, text: str, add_special_tokens: bool):
        tok = self._load(model_name)
        enc = tok(text, add_special_tokens=add_special_tokens,
                  return_offsets_mapping=True)
        return enc["input_ids"], enc.get("offset_mapping") or []

    def render_chat(self, model_name: str, messages, add_generation_prompt: bool):
        tok = self._load(model_name)
        rendered = tok.apply_chat_template(
            messages, tokenize=False, add_generation_prompt=add_generation_prompt
        )
        ids = tok(rendered, add_special_tokens=False)["input_ids"]
        return ids, rendered

    IMAGE_MARKER = "<image>"

    def render_chat_completion(self, model_name: str, messages,
                               add_generation_prompt: bool = True,
                               continue_final_message: bool = False,
                               chat_template: str = "",
                               chat_template_kwargs: str = ""):
        """OpenAI chat-completion render (reference RenderChatCompletion,
        tokenizer.proto:152-171): multimodal content parts are flattened —
        each image_url part becomes an IMAGE_MARKER in the text whose
        token span is reported as its placeholder range, and its mm hash
        is the sha256 of the image reference (content-addressed, matching
        what the engine's extra-keys carry). Text-only requests behave
        like render_chat."""
        import hashlib
        import json as _json

        tok = self._load(model_name)
        flat = []
        mm_hashes = []
        for m in messages:
            parts = m.get("content_parts")
            if parts:
                pieces = []
                for part in parts:
                    if part.get("type") == "image_url":
                        url = (part.get("image_url") or {}).get("url", "")
                        mm_hashes.append(
                            hashlib.sha256(url.encode()).hexdigest()[:32])
                        pieces.append(self.IMAGE_MARKER)
                    else:
                        pieces.append(part.get("text") or "")
                flat.append({"role": m["role"], "content": "".join(pieces)})
            else:
                flat.append({"role": m["role"],
                             "content": m.get("content") or ""})
        kwargs = {}
        if chat_template_kwargs:
            kwargs = _json.loads(chat_template_kwargs)
        rendered = tok.apply_chat_template(
            flat, tokenize=False,
            add_generation_prompt=add_generation_prompt,
            continue_final_message=continue_final_message,
            **({"chat_template": chat_template} if chat_template else {}),
            **kwargs)
        enc = tok(rendered, add_special_tokens=False,
                  return_offsets_mapping=True)
        ids = enc["input_ids"]
        offsets = enc.get("offset_mapping") or []
        ranges = []
        start = 0
        for _ in range(len(mm_hashes)):
            pos = rendered.find(self.IMAGE_MARKER, start)
            if pos < 0:
                break
            end = pos + len(self.IMAGE_MARKER)
            span = [i for i, (a, b) in enumerate(offsets)
                    if a < end and b > pos]
            if span:
                ranges.append((span[0], span[-1] - span[0] + 1))
            start = end
        return ids, mm_hashes, ranges

    def render_completion(self, model_name: str, prompt: str):
        """OpenAI completion render (reference RenderCompletion): plain
        prompt validation + tokenization."""
        tok = self._load(model_name)
        return tok(prompt, add_special_tokens=True)["input_ids"]

    async def run(self, fn, *args):
        return await asyncio.get_running_loop().run_in_executor(
            self._pool, fn, *args)


class TokenizationServicer:
    def __init__(self, manager: TokenizerManager):
        self.m = manager

    async def initialize(self, req, ctx):
        Resp = proto.get("tokenizerpb.InitializeTokenizerResponse")
        try:
            if req.tokenizer_path:
                self.m.register(req.model_name, req.tokenizer_path)
            await self.m.run(self.m._load, req.model_name)
            return Resp(success=True)
        except Exception as e:
            return Resp(success=False, error=str(e))

    async def tokenize(self, req, ctx):
        Resp = proto.get("tokenizerpb.TokenizeResponse")
        Off = proto.get("tokenizerpb.OffsetMapping")
        try:
            ids, offsets = await self.m.run(
                self.m.tokenize, req.model_name, req.text, req.add_special_tokens)
            return Resp(
                token_ids=ids,
                offsets=[Off(start=s, end=e) for s, e in offsets],
            )
        except Exception as e:
            log.warning("tokenize failed: %s", e)
            return Resp(error=str(e))

    async def render_chat(self, req, ctx):
        Resp = proto.get("tokenizerpb.RenderChatResponse")
        MM = proto.get("tokenizerpb.MultiModalFeatures")
        try:
            messages = [{"role": m.role, "content": m.content}
                        for m in req.messages]
            ids, rendered = await self.m.run(
                self.m.render_chat, req.model_name, messages,
                req.add_generation_prompt)
            resp = Resp(token_ids=ids, rendered=rendered)
            if req.mm_item_hashes:
                resp.mm_features.CopyFrom(MM(mm_hashes=list(req.mm_item_hashes)))
            return resp
        except Exception as e:
            log.warning("render_chat failed: %s", e)
            return Resp(error=str(e))

    async def render_chat_completion(self, req, ctx):
        import uuid

        Resp = proto.get("tokenizerpb.RenderChatCompletionResponse")
        MM = proto.get("tokenizerpb.MultiModalFeatures")
        PR = proto.get("tokenizerpb.PlaceholderRange")
        try:
            messages = []
            for m in req.messages:
                d = {"role": m.role, "content": m.content}
                if m.content_parts:
                    d["content_parts"] = [
                        {"type": p.type, "text": p.text,
                         "image_url": {"url": p.image_url.url}}
                        for p in m.content_parts
                    ]
                messages.append(d)
            ids, hashes, ranges = await self.m.run(
                lambda: self.m.render_chat_completion(
                    req.model_name, messages,
                    not req.HasField("add_generation_prompt")
                    or req.add_generation_prompt,
                    req.continue_final_message, req.chat_template,
                    req.chat_template_kwargs))
            resp = Resp(request_id=f"render-{uuid.uuid4().hex[:12]}",
                        token_ids=ids)
            if hashes:
                resp.features.CopyFrom(MM(
                    mm_hashes=hashes,
                    placeholder_ranges=[PR(offset=o, length=n)
                                        for o, n in ranges]))
            return resp
        except Exception as e:
            log.warning("render_chat_completion failed: %s", e)
            return Resp(error=str(e))

    async def render_completion(self, req, ctx):
        import uuid

        Resp = proto.get("tokenizerpb.RenderCompletionResponse")
        try:
            ids = await self.m.run(self.m.render_completion, req.model_name,
                                   req.prompt)
            return Resp(request_id=f"render-{uuid.uuid4().hex[:12]}",
                        token_ids=ids)
        except Exception as e:
            log.warning("render_completion failed: %s", e)
            return Resp(error=str(e))


def _handlers(servicer: TokenizationServicer):
    g = proto.get
    rpcs = {
        "InitializeTokenizer": (
            servicer.initialize,
            g("tokenizerpb.InitializeTokenizerRequest"),
            g("tokenizerpb.InitializeTokenizerResponse"),
        ),
        "Tokenize": (
            servicer.tokenize,
            g("tokenizerpb.TokenizeRequest"),
            g("tokenizerpb.TokenizeResponse"),
        ),
        "RenderChatTemplate": (
            servicer.render_chat,
            g("tokenizerpb.RenderChatRequest"),
            g("tokenizerpb.RenderChatResponse"),
        ),
        "RenderChatCompletion": (
            servicer.render_chat_completion,
            g("tokenizerpb.RenderChatCompletionRequest"),
            g("tokenizerpb.RenderChatCompletionResponse"),
        ),
        "RenderCompletion": (
            servicer.render_completion,
            g("tokenizerpb.RenderCompletionRequest"),
            g("tokenizerpb.RenderCompletionResponse"),
        ),
    }
    method_handlers = {}
    for name, (fn, Req, Resp) in rpcs.items():
        method_handlers[name] = grpc.unary_unary_rpc_method_handler(
            fn,
            request_deserializer=Req.FromString,
            response_serializer=Resp.SerializeToString,
        )
    return grpc.method_handlers_generic_handler(
        "tokenizerpb.TokenizationService", method_handlers)


async def serve(uds_path: str, manager: Optional[TokenizerManager] = None):
    """Start the sidecar; returns the grpc.aio server (caller stops it)."""
    manager = manager or TokenizerManager()
    server = grpc.aio.server(options=[
        ("grpc.max_receive_message_length", MAX_MSG),
        ("grpc.max_send_message_length", MAX_MSG),
    ])
    server.add_generic_rpc_handlers((_handlers(TokenizationServicer(manager)),))
    if os.path.exists(uds_path):
        os.unlink(uds_path)
    server.add_insecure_port(f"unix://{uds_path}")
    await server.start()
    return server


def main():  # pragma: no cover - operational entrypoint
    import argparse
    import json

    ap = argparse.ArgumentParser()
    ap.add_argument("--uds", default="/tmp/kvcache_tokenizer.sock")
    ap.add_argument("--model-paths-json", default=None,
                    help="JSON map model name -> local tokenizer dir")
    args = ap.parse_args()
    paths = json.loads(args.model_paths_json) if args.model_paths_json else {}

    async def _run():
        server = await serve(args.uds, TokenizerManager(paths))
        await server.wait_for_termination()

    asyncio.run(_run())


if __name__ == "__main__":  # pragma: no cover
    main()
