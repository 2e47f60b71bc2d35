"""Jupyter/IPython helpers (parity: torchx/notebook.py): a
``%%workspacefile`` cell magic that writes the cell body into an
in-memory fsspec workspace, so notebook users can assemble a workspace
and submit it with ``runner.run(..., workspace=get_workspace())``.

IPython is optional — importing this module without it raises only when
the magic is registered.
"""

from __future__ import annotations

import posixpath


def get_workspace() -> str:
    """The notebook workspace fsspec path (an in-memory filesystem)."""
    return "memory://torchx-workspace/"


def write_workspace_file(rel_path: str, content: str) -> str:
    """Write ``content`` at ``rel_path`` inside the notebook workspace."""
    import fsspec

    path = posixpath.join(get_workspace(), rel_path)
    fs, fpath = fsspec.core.url_to_fs(path)
    parent = posixpath.dirname(fpath)
    if parent:
        fs.makedirs(parent, exist_ok=True)
    with fs.open(fpath, "w") as f:
        f.write(content)
    return path


def register_magics() -> None:
    """Register the ``%%workspacefile <rel_path>`` cell magic (requires
    IPython; call from a notebook)."""
    from IPython.core.magic import register_cell_magic

    @register_cell_magic
    def workspacefile(line: str, cell: str) -> None:
        rel = line.strip()
        if not rel:
            raise ValueError(
                "usage: %%workspacefile <relative/path/in/workspace>"
            )
        write_workspace_file(rel, cell)


try:  # auto-register when imported inside IPython
    get_ipython()  # type: ignore[name-defined]  # noqa: F821
    register_magics()
except NameError:
    pass
