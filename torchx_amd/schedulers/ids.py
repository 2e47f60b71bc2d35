"""Unique job-id generation (parity: torchx/schedulers/ids.py)."""

from __future__ import annotations

import os
import re
import string

_ALPHABET = string.ascii_lowercase + string.digits


def random_id(length: int = 8) -> str:
    rnd = os.urandom(length)
    return "".join(_ALPHABET[b % len(_ALPHABET)] for b in rnd)


def make_unique(name: str) -> str:
    """``name`` -> ``name-ab12cd34`` (sanitized)."""
    safe = re.sub(r"[^a-zA-Z0-9\-_]", "-", name).strip("-") or "app"
    return f"{safe}-{random_id()}"
