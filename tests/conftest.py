import os
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an AMD GPU (MI355X)")
    # Build the native extensions up front so importing the package inside
    # tests never races.
    from llm_d_kv_cache_amd._build import build_all

    build_all()


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no AMD GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
