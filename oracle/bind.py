"""ctypes binding for the CPU oracle (liboracle.so) — TEST INFRASTRUCTURE ONLY.

Used by tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg as the
parity checker / reported CPU baseline; never on the product path.

Entry layouts match include/sre.h (sre_account_entry 104 B packed,
sre_storage_entry 96 B) so numpy structured arrays can be shared with the
product binding (reth_amd.engine).
"""
import ctypes
import os
import subprocess

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
LIB = os.path.join(HERE, "liboracle.so")

ACCOUNT_DTYPE = np.dtype([
    ("key", np.uint8, 32),
    ("nonce", np.uint64),
    ("balance", np.uint8, 32),
    ("code_hash", np.uint8, 32),
])  # 104 bytes, matches sre_account_entry
assert ACCOUNT_DTYPE.itemsize == 104

STORAGE_DTYPE = np.dtype([
    ("acct_key", np.uint8, 32),
    ("slot_key", np.uint8, 32),
    ("value", np.uint8, 32),
])  # 96 bytes, matches sre_storage_entry
assert STORAGE_DTYPE.itemsize == 96

UPDATE_DTYPE = np.dtype([
    ("acct_key", np.uint8, 32),
    ("kind", np.uint8),
    ("path_len", np.uint8),
    ("path", np.uint8, 32),
    ("num_hashes", np.uint8),
    ("root_hash_set", np.uint8),
    ("state_mask", "<u2"),
    ("tree_mask", "<u2"),
    ("hash_mask", "<u2"),
    ("root_hash", np.uint8, 32),
    ("hashes", np.uint8, (16, 32)),
    ("removed", np.uint8),  # oracle always emits 0 (full rebuilds)
    ("pad", np.uint8, 5),
])  # 624 bytes, matches sre_update_row
assert UPDATE_DTYPE.itemsize == 624

_lib = None


def build():
    subprocess.run(["make", "-s", "-C", HERE, "liboracle.so"], check=True)


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(LIB):
            build()
        _lib = ctypes.CDLL(LIB)
        _lib.okc_error_str.restype = ctypes.c_char_p
    return _lib


def _check(rc):
    if rc != 0:
        raise ValueError(f"oracle error {rc}: {lib().okc_error_str(rc).decode()}")


def _ptr(arr):
    if len(arr) == 0:
        return None
    return arr.ctypes.data_as(ctypes.c_void_p)


def keccak256(data: bytes) -> bytes:
    out = (ctypes.c_uint8 * 32)()
    lib().okc_keccak256(bytes(data), len(data), out)
    return bytes(out)


def keccak256_batch(msgs: np.ndarray) -> np.ndarray:
    """msgs: (n, L) uint8 array -> (n, 32) digests."""
    msgs = np.ascontiguousarray(msgs, dtype=np.uint8)
    n, length = msgs.shape
    out = np.empty((n, 32), dtype=np.uint8)
    lib().okc_keccak256_batch(_ptr(msgs), length, length, n, _ptr(out))
    return out


def state_root(accounts: np.ndarray, storage: np.ndarray) -> bytes:
    assert accounts.dtype == ACCOUNT_DTYPE and storage.dtype == STORAGE_DTYPE
    out = (ctypes.c_uint8 * 32)()
    _check(lib().okc_state_root(_ptr(accounts), len(accounts),
                                _ptr(storage), len(storage), out))
    return bytes(out)


def state_root_par(accounts: np.ndarray, storage: np.ndarray,
                   nthreads: int = 0) -> bytes:
    """Multi-threaded (OpenMP) state root — the one-thread-per-core CPU
    baseline leg of bench.py. nthreads 0 = all cores."""
    assert accounts.dtype == ACCOUNT_DTYPE and storage.dtype == STORAGE_DTYPE
    out = (ctypes.c_uint8 * 32)()
    _check(lib().okc_state_root_par(_ptr(accounts), len(accounts),
                                    _ptr(storage), len(storage),
                                    ctypes.c_int(nthreads), out))
    return bytes(out)


def storage_roots(accounts: np.ndarray, storage: np.ndarray) -> np.ndarray:
    out = np.empty((len(accounts), 32), dtype=np.uint8)
    _check(lib().okc_storage_roots(_ptr(accounts), len(accounts),
                                   _ptr(storage), len(storage), _ptr(out)))
    return out


def subtree_roots(accounts: np.ndarray, storage: np.ndarray):
    refs = np.zeros((16, 33), dtype=np.uint8)
    lens = np.zeros(16, dtype=np.uint8)
    roots = np.zeros((16, 32), dtype=np.uint8)
    counts = np.zeros(16, dtype=np.uint64)
    _check(lib().okc_subtree_roots(_ptr(accounts), len(accounts),
                                   _ptr(storage), len(storage),
                                   _ptr(refs), _ptr(lens), _ptr(roots),
                                   _ptr(counts)))
    return refs, lens, roots, counts


def state_root_with_updates(accounts: np.ndarray, storage: np.ndarray):
    """Returns (root_bytes, rows ndarray of UPDATE_DTYPE) — the stored
    trie nodes (TrieUpdates) of a full rebuild."""
    assert accounts.dtype == ACCOUNT_DTYPE and storage.dtype == STORAGE_DTYPE
    out = (ctypes.c_uint8 * 32)()
    rows_p = ctypes.c_void_p()
    n_rows = ctypes.c_uint64()
    _check(lib().okc_state_root_with_updates(
        _ptr(accounts), len(accounts), _ptr(storage), len(storage), out,
        ctypes.byref(rows_p), ctypes.byref(n_rows)))
    n = n_rows.value
    if n:
        buf = ctypes.string_at(rows_p.value, n * UPDATE_DTYPE.itemsize)
        rows = np.frombuffer(buf, dtype=UPDATE_DTYPE).copy()
        lib().okc_free_updates(rows_p)
    else:
        rows = np.zeros(0, dtype=UPDATE_DTYPE)
    return bytes(out), rows


def finish_top(refs, lens, roots, counts) -> bytes:
    out = (ctypes.c_uint8 * 32)()
    _check(lib().okc_finish_top(
        _ptr(np.ascontiguousarray(refs, dtype=np.uint8)),
        _ptr(np.ascontiguousarray(lens, dtype=np.uint8)),
        _ptr(np.ascontiguousarray(roots, dtype=np.uint8)),
        _ptr(np.ascontiguousarray(counts, dtype=np.uint64)), out))
    return bytes(out)
