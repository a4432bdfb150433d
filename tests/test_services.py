"""gRPC services: UDS tokenizer sidecar + indexer service.

Mirrors the reference e2e strategy (tests/e2e/uds_tokenizer) without
containers: a real grpc.aio UDS server with a toy HF tokenizer built
in-process (no network; the image has no downloaded models).
"""
import asyncio
import json
import os
import threading

import pytest

from llm_d_kv_cache_amd import ensure_native
from llm_d_kv_cache_amd.core import IndexerConfig, KVCacheIndexer

k = ensure_native()


@pytest.fixture(scope="module")
def toy_tokenizer_dir(tmp_path_factory):
    """Build a tiny whitespace WordLevel tokenizer + chat template."""
    d = tmp_path_factory.mktemp("toy_tok")
    from tokenizers import Tokenizer, models, pre_tokenizers

    vocab = {"<unk>": 0, "<s>": 1, "</s>": 2}
    for i, w in enumerate(
        "hello world the quick brown fox jumps over lazy dog user assistant "
        "system how are you i am fine".split()
    ):
        vocab[w] = i + 3
    tok = Tokenizer(models.WordLevel(vocab, unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    tok.save(str(d / "tokenizer.json"))
    cfg = {
        "tokenizer_class": "PreTrainedTokenizerFast",
        "model_max_length": 4096,
        "chat_template": (
            "{% for message in messages %}{{ message['role'] }} : "
            "{{ message['content'] }}\n{% endfor %}"
            "{% if add_generation_prompt %}assistant :{% endif %}"
        ),
        "unk_token": "<unk>", "bos_token": "<s>", "eos_token": "</s>",
    }
    (d / "tokenizer_config.json").write_text(json.dumps(cfg))
    return str(d)


@pytest.fixture(scope="module")
def uds_server(toy_tokenizer_dir, tmp_path_factory):
    """Run the async UDS sidecar on a dedicated event-loop thread."""
    from llm_d_kv_cache_amd.services.tokenizer_service import (
        TokenizerManager,
        serve,
    )

    sock = str(tmp_path_factory.mktemp("uds") / "tok.sock")
    loop = asyncio.new_event_loop()
    started = threading.Event()
    holder = {}

    def run():
        asyncio.set_event_loop(loop)

        async def go():
            mgr = TokenizerManager({"toy-model": toy_tokenizer_dir})
            holder["server"] = await serve(sock, mgr)
            started.set()
            await holder["server"].wait_for_termination()

        loop.run_until_complete(go())

    t = threading.Thread(target=run, daemon=True)
    t.start()
    assert started.wait(timeout=30)
    yield sock
    # fire-and-forget: the loop runs on a daemon thread; a blocking join can
    # deadlock against still-open client channels
    asyncio.run_coroutine_threadsafe(holder["server"].stop(0.1), loop)


def test_tokenize_roundtrip(uds_server):
    from llm_d_kv_cache_amd.services.tokenizer_client import UdsTokenizerClient

    c = UdsTokenizerClient(uds_server)
    c.initialize("toy-model")
    ids = c.tokenize("toy-model", "hello world", add_special_tokens=False)
    assert len(ids) == 2
    ids2 = c.tokenize("toy-model", "hello world hello", add_special_tokens=False)
    assert ids2[:2] == ids and ids2[2] == ids[0]
    c.close()


def test_tokenize_unknown_model_errors(uds_server):
    from llm_d_kv_cache_amd.services.tokenizer_client import UdsTokenizerClient

    c = UdsTokenizerClient(uds_server)
    with pytest.raises(RuntimeError):
        c.tokenize("no-such-model", "hello")
    c.close()


def test_render_chat_template(uds_server):
    from llm_d_kv_cache_amd.services.tokenizer_client import UdsTokenizerClient

    c = UdsTokenizerClient(uds_server)
    ids, rendered = c.render_chat(
        "toy-model",
        [("user", "hello world"), ("assistant", "how are you")],
        add_generation_prompt=True,
    )
    assert "user : hello world" in rendered
    assert rendered.endswith("assistant :")
    assert len(ids) > 4
    c.close()


def test_tokenization_pool_retries(uds_server):
    from llm_d_kv_cache_amd.services.tokenizer_client import (
        TokenizationPool,
        UdsTokenizerClient,
    )

    pool = TokenizationPool(UdsTokenizerClient(uds_server), workers=3)
    futs = [pool.tokenize_async("toy-model", "the quick brown fox")
            for _ in range(10)]
    for f in futs:
        assert len(f.result(timeout=30)) == 4
    pool.shutdown()


def test_indexer_service_score_tokens(uds_server):
    from llm_d_kv_cache_amd.services.indexer_service import (
        IndexerClient,
        create_server,
    )

    ix = KVCacheIndexer(IndexerConfig())
    tokens = list(range(32))
    keys = ix.compute_block_keys(tokens, "toy-model")
    ix.index.add([], keys, [k.PodEntry("pod-a", "gpu")])

    server, port = create_server(ix, "127.0.0.1:0")
    server.start()
    try:
        client = IndexerClient(f"127.0.0.1:{port}")
        scores = client.score_tokens(tokens, "toy-model")
        assert scores == {"pod-a": 2.0}
        assert client.score_tokens(tokens, "toy-model", pods=["nope"]) == {}
        client.close()
    finally:
        server.stop(0.1)


def test_indexer_service_prompt_path(uds_server):
    """Full prompt path: gRPC -> UDS tokenizer -> hash chain -> score."""
    from llm_d_kv_cache_amd.services.indexer_service import (
        IndexerClient,
        create_server,
    )
    from llm_d_kv_cache_amd.services.tokenizer_client import (
        TokenizationPool,
        UdsTokenizerClient,
    )

    tok_client = UdsTokenizerClient(uds_server)
    pool = TokenizationPool(tok_client)
    ix = KVCacheIndexer(IndexerConfig(
        token_processor=__import__(
            "llm_d_kv_cache_amd.core", fromlist=["TokenProcessorConfig"]
        ).TokenProcessorConfig(block_size_tokens=4),
    ))
    prompt = "the quick brown fox jumps over the lazy dog the quick brown"
    tokens = tok_client.tokenize("toy-model", prompt)
    keys = ix.compute_block_keys(tokens, "toy-model")
    assert keys, "prompt must span at least one block"
    ix.index.add([], keys, [k.PodEntry("pod-z", "gpu")])

    server, port = create_server(ix, "127.0.0.1:0", tokenizer_pool=pool)
    server.start()
    try:
        client = IndexerClient(f"127.0.0.1:{port}")
        scores = client.get_pod_scores(prompt, "toy-model")
        assert scores == {"pod-z": float(len(keys))}
        client.close()
    finally:
        server.stop(0.1)
        pool.shutdown()


def test_render_chat_completion_and_completion(uds_server):
    """OpenAI-shape render RPCs (reference RenderChatCompletion /
    RenderCompletion): multimodal parts produce content-addressed mm
    hashes + placeholder token ranges; plain completion tokenizes."""
    from llm_d_kv_cache_amd.services.tokenizer_client import (
        UdsTokenizerClient,
    )

    c = UdsTokenizerClient(uds_server)
    rid, ids, hashes, ranges = c.render_chat_completion(
        "toy-model",
        [{"role": "user", "content_parts": [
            {"type": "text", "text": "hello"},
            {"type": "image_url", "image_url": {"url": "data:img-1"}},
            {"type": "text", "text": "world"}]},
         {"role": "assistant", "content": "how are you"}],
    )
    assert rid.startswith("render-")
    assert len(ids) > 4
    assert len(hashes) == 1 and len(hashes[0]) == 32
    assert len(ranges) == 1 and ranges[0][1] >= 1
    # same image -> same hash (content-addressed); different -> different
    _, _, h2, _ = c.render_chat_completion(
        "toy-model",
        [{"role": "user", "content_parts": [
            {"type": "image_url", "image_url": {"url": "data:img-1"}},
            {"type": "image_url", "image_url": {"url": "data:img-2"}}]}],
    )
    assert h2[0] == hashes[0] and h2[1] != h2[0]

    # text-only chat completion matches the plain chat-template render
    ids_cc = c.render_chat_completion(
        "toy-model", [{"role": "user", "content": "hello world"}])[1]
    ids_rc = c.render_chat("toy-model", [("user", "hello world")],
                           add_generation_prompt=True)[0]
    assert ids_cc == ids_rc

    rid2, ids2 = c.render_completion("toy-model", "the quick brown fox")
    assert rid2.startswith("render-")
    assert len(ids2) == 4
    c.close()
