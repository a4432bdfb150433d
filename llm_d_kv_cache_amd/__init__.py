"""llm-d-kv-cache-amd: MI355X-native KV-cache management framework.

A from-scratch, AMD-native re-design of llm-d's KV-cache layer:

- ``core``     — global KV-block index + prefix-aware pod scoring
                 (native C++ ``_kvcore``: FNV-64a/CBOR hash chains, sharded
                 LRU index, longest-prefix scorer).
- ``events``   — KVEvents ingestion: from-scratch ZMTP 3.x PUB/SUB,
                 msgpack engine adapters (vLLM/SGLang), sharded worker pool.
- ``offload``  — KV-block data plane: CDNA4 (gfx950) HIP gather/scatter
                 kernels, NUMA-pinned async I/O threads, HBM -> pinned host
                 DRAM -> filesystem tiering (native ``_kvoffload``).
- ``peer``     — cross-GPU block migration over RCCL/xGMI.
- ``services`` — indexer gRPC service + UDS tokenizer sidecar.
- ``evictor``  — storage-tier disk-space manager.
"""
from __future__ import annotations

__version__ = "0.1.0"


def ensure_native(build_if_missing: bool = True):
    """Import (building if necessary) the native _kvcore module."""
    try:
        from . import _kvcore  # type: ignore[attr-defined]

        return _kvcore
    except ImportError:
        if not build_if_missing:
            raise
        from ._build import build_kvcore

        build_kvcore()
        from . import _kvcore  # type: ignore[attr-defined]

        return _kvcore


def ensure_offload_native(build_if_missing: bool = True):
    """Import (building if necessary) the native _kvoffload module.

    torch MUST be imported first: _kvoffload links libamdhip64, and if it
    loads the system copy before torch loads its bundled one, two HIP
    runtimes coexist and device enumeration fails ("no ROCm-capable device
    is detected") — observed on MI355X. Importing torch first makes the
    loader reuse torch's runtime for our extension.
    """
    import torch  # noqa: F401  (load order matters; see docstring)

    try:
        from . import _kvoffload  # type: ignore[attr-defined]

        return _kvoffload
    except ImportError:
        if not build_if_missing:
            raise
        from ._build import build_kvoffload

        build_kvoffload()
        from . import _kvoffload  # type: ignore[attr-defined]

        return _kvoffload
