"""C-ABI surface check (no GPU): the built libsre.so must load and export
every entry point include/sre.h declares — the driver's "does it build"
complement. No compute calls here (sre_create requires a HIP device and is
GPU-tested)."""
import ctypes
import os
import re

import pytest

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(HERE, "reth_amd", "libsre.so")
HDR = os.path.join(HERE, "include", "sre.h")


def _declared_symbols():
    text = open(HDR).read()
    # function declarations: "int sre_xxx(" / "sre_ctx *sre_create(" / etc.
    syms = set(re.findall(r"\b(sre_[a-z0-9_]+)\s*\(", text))
    return sorted(syms)


@pytest.fixture(scope="module")
def lib():
    if not os.path.exists(LIB):
        import subprocess
        r = subprocess.run(["hipcc", "--offload-arch=gfx950", "-O3",
                            "-std=c++17", "-shared", "-fPIC",
                            os.path.join(HERE, "reth_amd", "csrc", "sre.hip"),
                            "-o", LIB], capture_output=True, text=True)
        assert r.returncode == 0, r.stderr[-2000:]
    return ctypes.CDLL(LIB)


def test_header_parses_and_lists_expected_surface():
    syms = _declared_symbols()
    for must in ["sre_create", "sre_destroy", "sre_upload_accounts",
                 "sre_upload_storage", "sre_root", "sre_root_with_updates",
                 "sre_updates_count", "sre_updates_get", "sre_storage_roots",
                 "sre_subtree_roots", "sre_finish_top", "sre_apply_delta",
                 "sre_root_retaining", "sre_incremental_root",
                 "sre_account_proof", "sre_storage_proof", "sre_get_stats",
                 "sre_last_error"]:
        assert must in syms, must


def test_every_declared_symbol_exported(lib):
    missing = [s for s in _declared_symbols()
               if not hasattr(lib, s)]
    assert not missing, f"libsre.so missing exports: {missing}"


def test_struct_sizes_match_header_contract():
    from reth_amd.engine import (ACCOUNT_DTYPE, STORAGE_DTYPE, DELTA_DTYPE,
                                 UPDATE_DTYPE)
    assert ACCOUNT_DTYPE.itemsize == 104
    assert STORAGE_DTYPE.itemsize == 96
    assert DELTA_DTYPE.itemsize == 112
    assert UPDATE_DTYPE.itemsize == 624
