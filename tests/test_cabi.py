"""C-ABI library: builds, loads, and exports every symbol declared in
include/arrow_spmm.h (no compute here — runs without a GPU)."""
import ctypes
import os
import re
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, 'arrow_matrix_amd', 'libarrowspmm.so')
HEADER = os.path.join(REPO, 'include', 'arrow_spmm.h')


@pytest.fixture(scope='module')
def lib():
    if not os.path.exists(LIB):
        subprocess.run(['make', '-C', os.path.join(REPO, 'arrow_matrix_amd', 'csrc')],
                       check=True, capture_output=True)
    return ctypes.CDLL(LIB)


def _declared_symbols():
    src = open(HEADER).read()
    # function declarations: "<ret> arrow_xyz(" at line starts
    return sorted(set(re.findall(r'\b(arrow_[a-z0-9_]+)\s*\(', src)))


def test_header_symbols_exported(lib):
    syms = _declared_symbols()
    assert len(syms) >= 10
    for s in syms:
        assert hasattr(lib, s), f"symbol {s} declared in arrow_spmm.h but not exported"


def test_abi_version(lib):
    lib.arrow_abi_version.restype = ctypes.c_int
    assert lib.arrow_abi_version() >= 1000


def test_last_error_returns_string(lib):
    lib.arrow_last_error.restype = ctypes.c_char_p
    assert isinstance(lib.arrow_last_error(), bytes)
