"""Offload connector wiring: one config -> mapper + engine + handlers +
manager + storage events.

Capability parity with the reference SharedStorageOffloadingSpec
(llmd_fs_backend/spec.py): derives blocks-per-file from the offloaded
chunk size and per-group block sizes (GCD hash granularity across hybrid
groups), validates parallel layout, writes the run manifest, and exposes
``get_manager()`` / ``get_handlers()``. The vLLM OffloadingSpec plugin
subclass registers only when vLLM is importable (this image ships
without it); the framework-native ``OffloadConnector`` below is the same
wiring without the vLLM dependency.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Optional, Sequence

from .engine import OffloadEngineConfig, TorchOffloadEngine
from .events import StorageEventPublisher
from .file_mapper import FileMapper, KVCacheLayoutConfig
from .handlers import GPUToStorageHandler, StorageToGPUHandler
from .manager import SharedStorageOffloadManager

DEFAULT_OFFLOADED_BLOCK_TOKENS = 256  # file granularity (reference spec.py:39)


@dataclass
class OffloadConnectorConfig:
    root: str = "/mnt/kvcache"
    layout: KVCacheLayoutConfig = field(default_factory=KVCacheLayoutConfig)
    engine: OffloadEngineConfig = field(default_factory=OffloadEngineConfig)
    offloaded_block_tokens: int = DEFAULT_OFFLOADED_BLOCK_TOKENS
    # per KV-cache group, its engine block size in tokens
    group_block_tokens: Sequence[int] = (16,)
    # optional KVEvents endpoint announcing the storage tier to the indexer
    events_endpoint: Optional[str] = None


class OffloadConnector:
    """Framework-native connector: owns the engine, handlers and manager."""

    def __init__(self, groups: Sequence[Sequence], config: OffloadConnectorConfig):
        self.config = config
        # hash granularity = GCD of group block sizes (reference spec.py:80-89)
        self.hash_block_tokens = math.gcd(*[int(b) for b in config.group_block_tokens])
        if config.offloaded_block_tokens % self.hash_block_tokens != 0:
            raise ValueError(
                "offloaded_block_tokens must be a multiple of the GCD of "
                "group block sizes"
            )
        blocks_per_file = [
            config.offloaded_block_tokens // int(b)
            for b in config.group_block_tokens
        ]
        if max(blocks_per_file) > 4096:
            raise ValueError(
                "offloaded chunk spans more than 4096 engine blocks")
        eng_cfg = config.engine
        eng_cfg.gpu_blocks_per_file = max(blocks_per_file)
        self.engine = TorchOffloadEngine(groups, eng_cfg)
        self.blocks_per_file = blocks_per_file

        layout = config.layout
        layout.offloaded_block_tokens = config.offloaded_block_tokens
        layout.kv_cache_groups = tuple(
            ("group", int(bt), int(geo["block_bytes"]))
            for bt, geo in zip(config.group_block_tokens, self.engine.group_geometry)
        )
        self.mapper = FileMapper(config.root, layout)
        self.mapper.write_run_config()

        self.publisher = None
        if config.events_endpoint:
            self.publisher = StorageEventPublisher(
                config.events_endpoint, layout.model,
                offloaded_block_tokens=config.offloaded_block_tokens,
            )
        self.manager = SharedStorageOffloadManager(
            self.mapper, num_groups=len(self.blocks_per_file),
            publisher=self.publisher,
        )
        self.store_handler = GPUToStorageHandler(
            self.engine, self.mapper, self.blocks_per_file,
            group_block_tokens=config.group_block_tokens)
        self.load_handler = StorageToGPUHandler(
            self.engine, self.mapper, self.blocks_per_file,
            group_block_tokens=config.group_block_tokens)

    def get_manager(self) -> SharedStorageOffloadManager:
        return self.manager

    def get_handlers(self):
        return self.store_handler, self.load_handler

    def close(self):
        if self.publisher is not None:
            self.publisher.close()


# ---- vLLM plugin (registered only when vLLM is present) ---------------------

try:  # pragma: no cover - exercised only inside a vLLM worker
    from vllm.v1.kv_offload.spec import OffloadingSpec as _VllmOffloadingSpec

    class SharedStorageOffloadingSpec(_VllmOffloadingSpec):
        """vLLM OffloadingConnector plugin backed by the MI355X-native
        engine. Mirrors the reference plugin's extra-config keys
        (shared_storage_path, offloaded_block_size, ...)."""

        def __init__(self, vllm_config, kv_cache_config=None):
            # newer vLLM passes (vllm_config, kv_cache_config); older takes
            # vllm_config only — support both base arities
            try:
                super().__init__(vllm_config, kv_cache_config)
            except TypeError:
                super().__init__(vllm_config)
            extra = self.extra_config
            parallel = vllm_config.parallel_config
            layout = KVCacheLayoutConfig(
                model=vllm_config.model_config.model,
                dtype=str(vllm_config.model_config.dtype),
                tp_size=parallel.tensor_parallel_size,
                pp_size=parallel.pipeline_parallel_size,
            )
            self._connector_config = OffloadConnectorConfig(
                root=extra.get("shared_storage_path", "/mnt/kvcache"),
                layout=layout,
                offloaded_block_tokens=int(
                    extra.get("offloaded_block_size",
                              DEFAULT_OFFLOADED_BLOCK_TOKENS)),
                events_endpoint=extra.get("events_endpoint"),
            )
            self._connector = None

        def get_manager(self):
            # scheduler side: stateless manager needs only the mapper
            cfg = self._connector_config
            mapper = FileMapper(cfg.root, cfg.layout)
            return SharedStorageOffloadManager(mapper)

        def get_handlers(self, kv_caches):
            groups = [list(kv_caches.values())]
            cfg = self._connector_config
            cfg.group_block_tokens = (self.gpu_block_size,)
            if groups[0] and not groups[0][0].is_cuda:
                cfg.engine.copy_path = "host"  # CPU-stubbed test harness
            self._connector = OffloadConnector(groups, cfg)
            return self._connector.get_handlers()

except ImportError:
    SharedStorageOffloadingSpec = None  # vLLM not installed in this image
