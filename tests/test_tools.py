"""The tools/ scripts at least parse and compile on CPU (bitrot guard —
they are exercised for real on GPU boxes, outside CI)."""

import glob
import os
import py_compile

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("path", sorted(
    glob.glob(os.path.join(ROOT, "tools", "*.py"))),
    ids=lambda p: os.path.basename(p))
def test_tool_compiles(path):
    py_compile.compile(path, doraise=True)
