"""The C-ABI library loads and exports every symbol include/rrdb_engine.h
declares (no compute calls — runs without a GPU)."""
import ctypes
import os
import re

import pytest

from conftest import HIP_SO, ORACLE_SO, REPO

HEADER = os.path.join(REPO, "include", "rrdb_engine.h")


def _declared_symbols():
    src = open(HEADER).read()
    # function declarations: "rettype rrdb_xxx(" at top level
    syms = set(re.findall(r"\b(rrdb_[a-z_0-9]+)\s*\(", src))
    return syms


def _check(so_path):
    lib = ctypes.CDLL(so_path)
    missing = [s for s in sorted(_declared_symbols()) if not hasattr(lib, s)]
    assert not missing, f"{so_path} missing: {missing}"


def test_oracle_exports_all():
    _check(ORACLE_SO)


def test_hip_lib_exports_all():
    if not os.path.exists(HIP_SO):
        pytest.skip("librrdb_hip.so not built (run __graft_entry__.build())")
    _check(HIP_SO)


def test_hip_lib_refuses_cpu_only_open():
    """product path fails loudly without a GPU (no silent CPU fallback)."""
    if not os.path.exists(HIP_SO):
        pytest.skip("librrdb_hip.so not built")
    import torch

    if torch.cuda.is_available():
        pytest.skip("GPU present; covered by gpu tests")
    lib = ctypes.CDLL(HIP_SO)
    lib.rrdb_open.restype = ctypes.c_void_p
    lib.rrdb_open.argtypes = [ctypes.c_int32] * 3
    assert lib.rrdb_open(1, 0, -1) in (None, 0)
    assert lib.rrdb_open(1, 0, 0) in (None, 0)


def test_backend_strings():
    lib = ctypes.CDLL(ORACLE_SO)
    lib.rrdb_backend.restype = ctypes.c_char_p
    assert lib.rrdb_backend() == b"oracle-cpu"
    if os.path.exists(HIP_SO):
        lib2 = ctypes.CDLL(HIP_SO)
        lib2.rrdb_backend.restype = ctypes.c_char_p
        assert lib2.rrdb_backend() == b"hip-gfx950"


def test_oracle_header_marks_test_only():
    src = open(os.path.join(REPO, "oracle", "rrdb_oracle.c")).read()
    assert "TEST INFRASTRUCTURE ONLY" in src
