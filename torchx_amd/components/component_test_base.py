"""Public test helper for component authors.

Parity: torchx/components/component_test_base.py:33-121 — the reference
ships a ``ComponentTestCase`` whose ``validate(module, fn)`` runs the
component through the custom-component resolution path and checks its
``--help`` parses, so component authors can unit-test their definitions
the way ``torchx run path/to/file.py:fn --help`` would exercise them.

Usage::

    from torchx_amd.components.component_test_base import ComponentTestCase
    from torchx_amd.components import utils

    class MyTest(ComponentTestCase):
        def test_echo(self):
            self.validate(utils, "echo")
"""

from __future__ import annotations

import os
import shutil
import tempfile
import unittest
from types import ModuleType

from torchx_amd.specs.builders import create_args_parser
from torchx_amd.specs.finder import get_component


class ComponentTestCase(unittest.TestCase):
    """TestCase with helpers for validating component definitions."""

    def setUp(self) -> None:
        self.test_dir = tempfile.mkdtemp("torchx_amd_component_test")
        self.old_cwd = os.getcwd()

    def tearDown(self) -> None:
        shutil.rmtree(self.test_dir, ignore_errors=True)
        os.chdir(self.old_cwd)

    def validate(self, module: ModuleType, function_name: str) -> None:
        """Resolve ``module.function_name`` as a custom component
        (``/abs/path/file.py:fn``) and check its ``--help`` exits cleanly —
        the component-author equivalent of
        ``torchx run file.py:fn --help``. Raises (failing the test) if the
        component does not lint, resolve, or parse."""
        module_path = module.__file__
        assert module_path, f"module must have __file__: {module}"
        component_id = f"{os.path.abspath(module_path)}:{function_name}"
        component_def = get_component(component_id)
        with self.assertRaises(SystemExit):
            create_args_parser(component_def.fn).parse_args(["--help"])
