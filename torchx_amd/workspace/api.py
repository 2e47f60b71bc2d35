"""Workspace mixin: overlay a local project dir onto the role image at
submit time (parity: torchx/workspace/api.py: WorkspaceMixin :79,
walk_workspace :198 with .torchxignore support)."""

from __future__ import annotations

import fnmatch
import os
import posixpath
import tempfile
from typing import Any, Iterable, List, Mapping, Tuple, Union

from torchx_amd.specs import AppDef, Role, Workspace

WorkspaceLike = Union[str, "Workspace"]

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


def merge_workspace(ws: "Workspace", outdir: str) -> None:
    """Copy every project of ``ws`` into ``outdir`` under its destination
    path, honoring each project's .torchxignore; later projects win on
    file conflicts (parity: torchx/workspace/api.py:149-154)."""
    import shutil

    for src, dst in ws.projects.items():
        base = os.path.join(outdir, dst) if dst else outdir
        for abs_path, rel in walk_workspace(src):
            target = os.path.join(base, rel)
            os.makedirs(os.path.dirname(target) or base, exist_ok=True)
            shutil.copy2(abs_path, target)


class WorkspaceMixin:
    """Schedulers mix this in to support workspace patching."""

    def workspace_opts(self):
        from torchx_amd.specs import runopts

        return runopts()

    def build_workspaces(self, app: AppDef, workspace: "WorkspaceLike",
                         cfg: Mapping[str, Any]) -> None:
        """Build each role's workspace and update ``role.image`` in place.

        ``workspace`` may be a plain dir, a ``"src:dst,src2:dst2"`` spec
        string, or a :class:`~torchx_amd.specs.Workspace`. Multi-project
        workspaces are merged into a tmpdir first; builds are cached per
        (image, workspace) so roles sharing both build once.
        """
        from torchx_amd.specs import Workspace

        ws = Workspace.from_str(workspace)
        if not ws:
            return
        build_cache: dict = {}
        key_ws = tuple(sorted(ws.projects.items()))
        merged_dir: str = ""
        tmp = None
        try:
            for role in app.roles:
                key = (role.image, key_ws)
                if key in build_cache:
                    role.image = build_cache[key]
                    continue
                if ws.is_unmapped_single_project():
                    build_dir = next(iter(ws.projects))
                else:
                    if not merged_dir:
                        tmp = tempfile.TemporaryDirectory(
                            suffix="_torchx_workspace")
                        merged_dir = tmp.name
                        merge_workspace(ws, merged_dir)
                    build_dir = merged_dir
                build_cache[key] = self.build_workspace_and_update_role(
                    role, build_dir, cfg
                )
        finally:
            if tmp is not None:
                tmp.cleanup()

    def build_workspace_and_update_role(self, role: Role, workspace: str,
                                        cfg: Mapping[str, Any]) -> str:
        raise NotImplementedError

    def dryrun_push_images(self, app: AppDef, cfg: Mapping[str, Any]) -> Any:
        return None

    def push_images(self, images: Any) -> None:
        pass
