"""KV-block offload data plane: GPU HBM -> pinned host DRAM -> filesystem.

Python layer over the native ``_kvoffload`` engine (HIP/CDNA4). Capability
parity with the reference ``kv_connectors/llmd_fs_backend``:

- ``file_mapper``   content-addressed storage layout
- ``engine``        torch -> native engine construction
- ``handlers``      transfer building (per-file splits, partial chunks)
- ``manager``       scheduler-side lookup / prepare / complete + events
- ``events``        storage-tier KVEvents publisher (feeds the indexer)
"""
from .engine import OffloadEngineConfig, TorchOffloadEngine  # noqa: F401
from .file_mapper import FileMapper, KVCacheLayoutConfig  # noqa: F401
from .handlers import GPUToStorageHandler, StorageToGPUHandler, TransferResult  # noqa: F401
from .manager import SharedStorageOffloadManager  # noqa: F401
