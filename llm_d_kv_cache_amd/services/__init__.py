"""gRPC services: indexer scoring API + UDS tokenizer sidecar."""
from .indexer_service import IndexerClient, create_server  # noqa: F401
from .tokenizer_client import TokenizationPool, UdsTokenizerClient  # noqa: F401
from .tokenizer_service import TokenizerManager, serve  # noqa: F401
