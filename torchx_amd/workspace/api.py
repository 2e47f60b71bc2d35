"""Workspace mixin: overlay a local project dir onto the role image at
submit time (parity: torchx/workspace/api.py: WorkspaceMixin :79,
walk_workspace :198 with .torchxignore support)."""

from __future__ import annotations

import fnmatch
import os
import posixpath
from typing import Any, Iterable, List, Mapping, Tuple

from torchx_amd.specs import AppDef, Role

TORCHX_IGNORE = ".torchxignore"


def _load_ignore_patterns(workspace: str) -> List[Tuple[bool, str]]:
    """Returns (negated, pattern) pairs from .torchxignore/.dockerignore."""
    patterns: List[Tuple[bool, str]] = []
    for fname in (TORCHX_IGNORE, ".dockerignore"):
        path = os.path.join(workspace, fname)
        if os.path.isfile(path):
            with open(path) as f:
                for line in f:
                    line = line.strip()
                    if not line or line.startswith("#"):
                        continue
                    neg = line.startswith("!")
                    patterns.append((neg, line[1:] if neg else line))
    return patterns


def _ignored(rel: str, patterns: List[Tuple[bool, str]]) -> bool:
    ignored = False
    for neg, pat in patterns:
        pat = pat.rstrip("/")
        if fnmatch.fnmatch(rel, pat) or fnmatch.fnmatch(rel, pat + "/*") or \
                rel.startswith(pat + "/"):
            ignored = not neg
    return ignored


def walk_workspace(workspace: str) -> Iterable[Tuple[str, str]]:
    """Yields (abs_path, rel_path) of files to include, honoring ignores."""
    patterns = _load_ignore_patterns(workspace)
    for root, dirs, files in os.walk(workspace):
        rel_root = os.path.relpath(root, workspace)
        if rel_root == ".":
            rel_root = ""
        dirs[:] = [
            d for d in dirs
            if not _ignored(posixpath.join(rel_root, d), patterns)
        ]
        for f in files:
            rel = posixpath.join(rel_root, f) if rel_root else f
            if not _ignored(rel, patterns):
                yield os.path.join(root, f), rel


class WorkspaceMixin:
    """Schedulers mix this in to support workspace patching."""

    def workspace_opts(self):
        from torchx_amd.specs import runopts

        return runopts()

    def build_workspaces(self, app: AppDef, workspace: str,
                         cfg: Mapping[str, Any]) -> None:
        images: dict = {}
        for role in app.roles:
            key = (role.image, workspace)
            if key not in images:
                images[key] = self.build_workspace_and_update_role(
                    role, workspace, cfg
                )
            else:
                role.image = images[key]
        for key, img in images.items():
            pass

    def build_workspace_and_update_role(self, role: Role, workspace: str,
                                        cfg: Mapping[str, Any]) -> str:
        raise NotImplementedError

    def dryrun_push_images(self, app: AppDef, cfg: Mapping[str, Any]) -> Any:
        return None

    def push_images(self, images: Any) -> None:
        pass
