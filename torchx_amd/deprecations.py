"""Deprecation shims (reference parity: torchx/deprecations.py:26,69).

``deprecated`` wraps a callable with a one-shot DeprecationWarning;
``deprecated_module`` emits one at import of a renamed/moved module.
"""

from __future__ import annotations

import functools
import warnings
from typing import Any, Callable, TypeVar

F = TypeVar("F", bound=Callable[..., Any])


def deprecated(replacement: str = "") -> Callable[[F], F]:
    """Mark a function deprecated; emits DeprecationWarning on first call."""

    def decorate(fn: F) -> F:
        warned = False

        @functools.wraps(fn)
        def wrapper(*args: Any, **kwargs: Any) -> Any:
            nonlocal warned
            if not warned:
                msg = f"{fn.__module__}.{fn.__qualname__} is deprecated"
                if replacement:
                    msg += f"; use {replacement} instead"
                warnings.warn(msg, DeprecationWarning, stacklevel=2)
                warned = True
            return fn(*args, **kwargs)

        return wrapper  # type: ignore[return-value]

    return decorate


def deprecated_module(old: str, new: str) -> None:
    """Call at the top of a moved module's compatibility shim."""
    warnings.warn(
        f"module {old} is deprecated; import {new} instead",
        DeprecationWarning,
        stacklevel=3,
    )
