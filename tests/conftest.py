import ctypes
import sys
from pathlib import Path

import pytest

# make the in-tree package importable regardless of cwd
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def is_asan() -> bool:
    """Interpreter built with ASAN (parity: the reference's sanitizer
    guards, torchx/schedulers/test/test_util.py:15-25 — subprocess-heavy
    tests misbehave under sanitizer interceptors)."""
    try:
        return hasattr(ctypes.CDLL(""), "__asan_init")
    except OSError:
        return False


def is_tsan() -> bool:
    try:
        return hasattr(ctypes.CDLL(""), "__tsan_init")
    except OSError:
        return False


def is_asan_or_tsan() -> bool:
    return is_asan() or is_tsan()


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run on the GPU box)"
    )
    config.addinivalue_line(
        "markers", "subprocess_heavy: skipped under ASAN/TSAN interpreters"
    )


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if not has_gpu:
        skip = pytest.mark.skip(reason="no GPU in this container")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)
    if is_asan_or_tsan():
        skip_san = pytest.mark.skip(
            reason="subprocess-heavy test under an ASAN/TSAN interpreter")
        for item in items:
            if "subprocess_heavy" in item.keywords:
                item.add_marker(skip_san)
