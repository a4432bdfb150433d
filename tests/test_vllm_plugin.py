"""vLLM OffloadingSpec plugin exercised against vendored stub classes.

The image ships no vLLM, so — exactly like the reference's CPU tests,
which hand-roll the vLLM classes they need (tests/cpu/
test_storage_events.py:15-50) — minimal stubs mirroring vLLM's
`OffloadingSpec` contract are injected into sys.modules and the plugin
subclass is constructed, wired, and driven through a store/load
round-trip. This keeps `SharedStorageOffloadingSpec` out of dead-code
territory in CI; real-engine validation still needs a vLLM image
(docs/limitations.md).
"""
import importlib
import sys
import time
import types
from contextlib import contextmanager
from types import SimpleNamespace

import pytest
import torch


class _StubOffloadingSpec:
    """Mirrors vllm.v1.kv_offload.spec.OffloadingSpec's observable
    contract: ctor(vllm_config, kv_cache_config) storing the configs and
    exposing extra_config + gpu_block_size."""

    def __init__(self, vllm_config, kv_cache_config=None):
        self.vllm_config = vllm_config
        self.kv_cache_config = kv_cache_config
        ktc = vllm_config.kv_transfer_config
        self.extra_config = ktc.kv_connector_extra_config
        self.gpu_block_size = vllm_config.cache_config.block_size
        self.offloaded_block_size = int(
            self.extra_config.get("block_size",
                                  vllm_config.cache_config.block_size))


@contextmanager
def vllm_stubs():
    mods = {}
    vllm = types.ModuleType("vllm")
    v1 = types.ModuleType("vllm.v1")
    kv_offload = types.ModuleType("vllm.v1.kv_offload")
    spec_mod = types.ModuleType("vllm.v1.kv_offload.spec")
    spec_mod.OffloadingSpec = _StubOffloadingSpec
    vllm.v1 = v1
    v1.kv_offload = kv_offload
    kv_offload.spec = spec_mod
    mods = {"vllm": vllm, "vllm.v1": v1, "vllm.v1.kv_offload": kv_offload,
            "vllm.v1.kv_offload.spec": spec_mod}
    sentinel = object()
    prev = {k: sys.modules.get(k, sentinel) for k in mods}
    sys.modules.update(mods)
    import llm_d_kv_cache_amd.offload.spec as our_spec
    importlib.reload(our_spec)
    try:
        yield our_spec
    finally:
        for k, v in prev.items():
            if v is sentinel:
                sys.modules.pop(k, None)
            else:
                sys.modules[k] = v
        importlib.reload(our_spec)


def make_vllm_config(tmp_path, block_size=16):
    return SimpleNamespace(
        model_config=SimpleNamespace(model="stub/model-8b", dtype="bfloat16"),
        parallel_config=SimpleNamespace(tensor_parallel_size=1,
                                        pipeline_parallel_size=1),
        cache_config=SimpleNamespace(block_size=block_size),
        kv_transfer_config=SimpleNamespace(kv_connector_extra_config={
            "shared_storage_path": str(tmp_path),
            "offloaded_block_size": 64,
        }),
    )


def test_plugin_registers_under_vllm():
    import llm_d_kv_cache_amd.offload.spec as our_spec

    assert our_spec.SharedStorageOffloadingSpec is None  # no vLLM here
    with vllm_stubs() as reloaded:
        assert reloaded.SharedStorageOffloadingSpec is not None
        assert issubclass(reloaded.SharedStorageOffloadingSpec,
                          _StubOffloadingSpec)
    importlib.invalidate_caches()
    import llm_d_kv_cache_amd.offload.spec as restored
    assert restored.SharedStorageOffloadingSpec is None


def test_plugin_manager_and_handlers_roundtrip(tmp_path):
    with vllm_stubs() as spec_mod:
        cfg = make_vllm_config(tmp_path)
        plugin = spec_mod.SharedStorageOffloadingSpec(
            cfg, kv_cache_config=SimpleNamespace())
        assert plugin.gpu_block_size == 16
        assert plugin._connector_config.offloaded_block_tokens == 64

        # scheduler side: stateless manager (lookup = file existence)
        mgr = plugin.get_manager()
        assert mgr.lookup([1, 2, 3]) == 0

        # worker side: handlers over vLLM-shaped kv_caches (layer -> tensor)
        kv_caches = {
            f"model.layers.{i}.self_attn": torch.randint(
                0, 255, (64, 4096), dtype=torch.uint8)
            for i in range(4)
        }
        store, load = plugin.get_handlers(kv_caches)
        assert store.blocks_per_file == [4]  # 64 offloaded / 16 gpu tokens

        ids = list(range(8))
        store.transfer_async([0xAB, 0xAC], {0: ids})
        deadline = time.time() + 15
        done = []
        while not done and time.time() < deadline:
            done = store.get_finished()
            time.sleep(0.005)
        assert done and done[0].success

        golden = [t[:8].clone() for t in kv_caches.values()]
        for t in kv_caches.values():
            t.zero_()
        load.transfer_async([0xAB, 0xAC], {0: ids})
        deadline = time.time() + 15
        done = []
        while not done and time.time() < deadline:
            done = load.get_finished()
            time.sleep(0.005)
        assert done and done[0].success
        for t, g in zip(kv_caches.values(), golden):
            assert torch.equal(t[:8], g)

        # the manager now sees the stored chunks through the same mapper
        mgr2 = plugin.get_manager()
        assert mgr2.lookup([0xAB, 0xAC, 0xFF]) == 2


def test_plugin_respects_extra_config_defaults(tmp_path):
    with vllm_stubs() as spec_mod:
        cfg = make_vllm_config(tmp_path)
        cfg.kv_transfer_config.kv_connector_extra_config = {}
        plugin = spec_mod.SharedStorageOffloadingSpec(cfg)
        cc = plugin._connector_config
        assert cc.root == "/mnt/kvcache"
        assert cc.offloaded_block_tokens == 256  # reference default
