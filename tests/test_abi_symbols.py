"""CPU test: libbifrost.so loads and exports every function that
include/bifrost/*.h declares (the C-ABI drop-in contract).
No compute calls — just symbol presence."""

import ctypes
import glob
import os
import re

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
INCLUDE = os.path.join(REPO, "include", "bifrost")
LIB = os.path.join(REPO, "bifrost_amd", "lib", "libbifrost.so")

# Matches 'BFstatus bfFoo(' / 'BFbool bfFoo(' / 'const char* bfFoo(' etc.
_DECL_RE = re.compile(
    r"^\s*(?:BFstatus|BFbool|BFsize|const\s+char\s*\*)\s+(bf[A-Za-z0-9_]+)\s*\(",
    re.MULTILINE)


def declared_symbols():
    syms = set()
    for path in sorted(glob.glob(os.path.join(INCLUDE, "*.h"))):
        with open(path) as f:
            syms.update(_DECL_RE.findall(f.read()))
    return syms


def test_headers_found():
    assert os.path.isdir(INCLUDE)
    assert len(declared_symbols()) >= 70  # SURVEY.md §8b lists ~80 entries


def test_library_loads():
    assert os.path.exists(LIB), "libbifrost.so not built (make -C bifrost_amd/csrc)"
    ctypes.CDLL(LIB)


@pytest.mark.parametrize("sym", sorted(declared_symbols()))
def test_symbol_exported(sym):
    lib = ctypes.CDLL(LIB)
    assert hasattr(lib, sym), "missing C-ABI symbol: %s" % sym
