"""Content-addressed storage layout for offloaded KV-block files.

Layout (capability parity with the reference file_mapper.py:112-143):

    <root>/<model>_<sha256(config)[:12]>_r<rank>/<h[0:3]>/<h[3:5]>_g<group>/<h>.bin

where ``h`` is the 16-hex-digit chunk hash. Every config field that changes
the bytes on disk (parallel sizes/ranks, dtype, group geometry) is folded
into the run directory hash so incompatible layouts never collide; the
``parallel_agnostic`` option collapses rank-independent layouts.
"""
from __future__ import annotations

import hashlib
import json
import os
import re
from dataclasses import asdict, dataclass
from typing import Tuple


@dataclass
class KVCacheLayoutConfig:
    model: str = "unknown"
    dtype: str = "bfloat16"
    tp_size: int = 1
    tp_rank: int = 0
    pp_size: int = 1
    pp_rank: int = 0
    # prefill/decode context parallel (long-context engines offload per rank)
    pcp_size: int = 1
    pcp_rank: int = 0
    dcp_size: int = 1
    dcp_rank: int = 0
    # per-group (kind, block_size_tokens, page_bytes) tuples — HMA geometry
    kv_cache_groups: Tuple = ()
    offloaded_block_tokens: int = 256
    parallel_agnostic: bool = False

    @property
    def world_rank(self) -> int:
        # rank order: pp outer, then tp, then pcp, then dcp
        r = self.pp_rank
        r = r * self.tp_size + self.tp_rank
        r = r * self.pcp_size + self.pcp_rank
        r = r * self.dcp_size + self.dcp_rank
        return r


class FileMapper:
    def __init__(self, root: str, config: KVCacheLayoutConfig):
        self.root = root
        self.config = config
        self.run_dir = os.path.join(root, self._run_dir_name())

    def _canonical_config(self) -> dict:
        cfg = asdict(self.config)
        cfg["kv_cache_groups"] = [list(g) for g in self.config.kv_cache_groups]
        if self.config.parallel_agnostic:
            for k in ("tp_size", "tp_rank", "pp_size", "pp_rank",
                      "pcp_size", "pcp_rank", "dcp_size", "dcp_rank"):
                cfg.pop(k)
        else:
            # ranks shape the path via _r<rank>, not the hash, so all ranks
            # of one run share a config hash
            for k in ("tp_rank", "pp_rank", "pcp_rank", "dcp_rank"):
                cfg.pop(k)
        return cfg

    def _run_dir_name(self) -> str:
        cfg = self._canonical_config()
        digest = hashlib.sha256(
            json.dumps(cfg, sort_keys=True).encode()
        ).hexdigest()[:12]
        model = re.sub(r"[^A-Za-z0-9_.-]", "_", self.config.model)
        rank = 0 if self.config.parallel_agnostic else self.config.world_rank
        return f"{model}_{digest}_r{rank}"

    def file_name(self, chunk_hash: int, group: int = 0) -> str:
        h = f"{chunk_hash & ((1 << 64) - 1):016x}"
        return os.path.join(self.run_dir, h[0:3], f"{h[3:5]}_g{group}", f"{h}.bin")

    def write_run_config(self) -> str:
        """Persist the run manifest (config.json) for restart compatibility
        checks and operator debugging."""
        os.makedirs(self.run_dir, exist_ok=True)
        path = os.path.join(self.run_dir, "config.json")
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(self._canonical_config(), f, sort_keys=True, indent=1)
        os.replace(tmp, path)
        return path
