"""AST linter for component functions.

Behavior parity with the reference (torchx/specs/file_linter.py:91-403):
a component function must annotate every argument with a supported type
(primitives, ``Optional``/``List``/``Dict`` of primitives, ``*args: str``),
declare an ``AppDef`` return type, and should carry a google-style docstring
describing each argument (missing docs are warnings, not errors).
"""

from __future__ import annotations

import argparse
import ast
from dataclasses import dataclass
from typing import List, Optional

_PRIMITIVES = {"int", "float", "str", "bool"}
_CONTAINERS = {"List", "Dict", "Optional", "Tuple", "list", "dict", "tuple"}


class ComponentHelpFormatter(argparse.RawDescriptionHelpFormatter):
    """Help formatter for component arg parsers: appends ``(required)`` to
    required arguments and ``(default: X)`` otherwise (parity:
    torchx/specs/file_linter.py:33-55 TorchXArgumentHelpFormatter —
    component functions never have an argument that is both)."""

    def _get_help_string(self, action: argparse.Action) -> str:
        help_txt = action.help or ""
        if action.default is argparse.SUPPRESS:  # only --help
            return help_txt
        if action.required:
            return f"{help_txt} (required)".strip()
        return f"{help_txt} (default: {action.default})".strip()


@dataclass
class LinterMessage:
    name: str
    description: str
    line: int
    char: int = 0
    severity: str = "error"


def _unparse(node: Optional[ast.expr]) -> str:
    return ast.unparse(node) if node is not None else ""


def _optional_inner(ann: ast.expr) -> Optional[ast.expr]:
    """The ``T`` of ``Optional[T]`` / ``T | None`` / ``None | T``, else None."""
    if (isinstance(ann, ast.Subscript) and isinstance(ann.value, ast.Name)
            and ann.value.id == "Optional"):
        return ann.slice
    if isinstance(ann, ast.BinOp) and isinstance(ann.op, ast.BitOr):
        if isinstance(ann.right, ast.Constant) and ann.right.value is None:
            return ann.left
        if isinstance(ann.left, ast.Constant) and ann.left.value is None:
            return ann.right
    return None


def _unwrap_annotated(ann: ast.expr) -> ast.expr:
    """``Annotated[T, meta...]`` -> ``T`` (metadata carries e.g. short
    flags; only the leading type is validated)."""
    if (isinstance(ann, ast.Subscript) and isinstance(ann.value, ast.Name)
            and ann.value.id == "Annotated"):
        sl = ann.slice
        if isinstance(sl, ast.Tuple) and sl.elts:
            return sl.elts[0]
    return ann


def _type_ok(ann: ast.expr, depth: int = 0) -> bool:
    ann = _unwrap_annotated(ann)
    inner = _optional_inner(ann)
    if inner is not None:
        return _type_ok(inner, depth)
    if isinstance(ann, ast.Name):
        return ann.id in _PRIMITIVES
    if isinstance(ann, ast.Subscript) and isinstance(ann.value, ast.Name):
        base = ann.value.id
        if base not in _CONTAINERS or depth > 0:
            return False
        sl = ann.slice
        elts = sl.elts if isinstance(sl, ast.Tuple) else [sl]
        if base in ("Dict", "dict") and len(elts) != 2:
            return False  # dict needs exactly K and V type params
        if base in ("Tuple", "tuple") and len(elts) < 2:
            return False  # single-element tuples unsupported for components
        return all(_type_ok(e, depth + 1) for e in elts)
    return False


class _Visitor(ast.NodeVisitor):
    def __init__(self, fn_name: str):
        self.fn_name = fn_name
        self.messages: List[LinterMessage] = []
        self.found = False

    def _err(self, desc: str, line: int, severity: str = "error") -> None:
        self.messages.append(
            LinterMessage("ComponentLinter", desc, line, severity=severity)
        )

    def visit_FunctionDef(self, node: ast.FunctionDef) -> None:
        if node.name != self.fn_name:
            return
        self.found = True
        args = list(node.args.args) + list(node.args.kwonlyargs)
        for arg in args:
            if arg.annotation is None:
                self._err(
                    f"missing type annotation for argument `{arg.arg}` "
                    f"in component `{node.name}`", arg.lineno)
            elif not _type_ok(arg.annotation):
                self._err(
                    f"unsupported type `{_unparse(arg.annotation)}` for "
                    f"argument `{arg.arg}` in component `{node.name}`: use a "
                    f"primitive, Optional/List/Dict of primitives", arg.lineno)
        va = node.args.vararg
        if va is not None and va.annotation is not None:
            if not (isinstance(va.annotation, ast.Name)
                    and va.annotation.id == "str"):
                self._err(
                    f"varargs `*{va.arg}` in component `{node.name}` must be "
                    f"`str`", va.lineno)
        ret = node.returns
        if ret is None or "AppDef" not in _unparse(ret):
            self._err(
                f"component `{node.name}` must declare `-> AppDef`",
                node.lineno)
        doc = ast.get_docstring(node)
        if not doc:
            self._err(
                f"component `{node.name}` has no docstring; add one with an "
                f"Args: section", node.lineno, severity="warning")


def validate(path: str, fn_name: str) -> List[LinterMessage]:
    """Lint the component function ``fn_name`` defined in the file ``path``."""
    with open(path, "r") as f:
        source = f.read()
    try:
        tree = ast.parse(source, filename=path)
    except SyntaxError as e:
        return [LinterMessage("ComponentLinter", str(e), e.lineno or 0)]
    v = _Visitor(fn_name)
    v.visit(tree)
    if not v.found:
        return [LinterMessage(
            "ComponentLinter", f"function `{fn_name}` not found in {path}", 0)]
    return v.messages
