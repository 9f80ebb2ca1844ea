"""ctypes binding over the C-ABI of the MI355X state-root engine (libsre.so).

Product path: GPU-only. Raises RuntimeError at construction when no HIP
device / extension is present — there is no CPU fallback (the CPU oracle
under oracle/ is test infrastructure and must never be imported here).

Mirrors the reference surface it replaces (see include/sre.h):
  StateRootEngine.root()           ~ StateRoot::root() /
                                     StateRootProvider::state_root
                                     (/root/reference/crates/storage/storage-api/src/trie.rs:13-41)
  StateRootEngine.storage_roots()  ~ StorageRootProvider::storage_root (:45-58)
  subtree_roots()/finish_top()     ~ the multi-GPU decomposition of
                                     StateRoot::calculate (crates/trie/trie/src/trie.rs:171)
"""
import ctypes
import os
import subprocess

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
LIB = os.path.join(HERE, "libsre.so")
SRC = os.path.join(HERE, "csrc", "sre.hip")

ACCOUNT_DTYPE = np.dtype([
    ("key", np.uint8, 32),
    ("nonce", np.uint64),
    ("balance", np.uint8, 32),
    ("code_hash", np.uint8, 32),
])
STORAGE_DTYPE = np.dtype([
    ("acct_key", np.uint8, 32),
    ("slot_key", np.uint8, 32),
    ("value", np.uint8, 32),
])

# sre_account_delta (include/sre.h): one overlay-delta account row
DELTA_DTYPE = np.dtype([
    ("key", np.uint8, 32),
    ("nonce", np.uint64),
    ("balance", np.uint8, 32),
    ("code_hash", np.uint8, 32),
    ("deleted", np.uint8),
    ("pad", np.uint8, 7),
])
assert DELTA_DTYPE.itemsize == 112

# sre_update_row (include/sre.h): one stored BranchNodeCompact + its path
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
    # incremental net-diff marker: 0 upsert, 1 removed path,
    # 2 whole-storage-trie deletion (destroyed account) — include/sre.h
    ("removed", np.uint8),
    ("pad", np.uint8, 5),
])
assert UPDATE_DTYPE.itemsize == 624


class SreStats(ctypes.Structure):
    _fields_ = [
        ("total_ms", ctypes.c_double),
        ("leaf_hash_ms", ctypes.c_double),
        ("leaf_count", ctypes.c_uint64),
        ("leaf_blocks", ctypes.c_uint64),
        ("branch_hash_ms", ctypes.c_double),
        ("branch_count", ctypes.c_uint64),
        ("branch_blocks", ctypes.c_uint64),
        ("sort_ms", ctypes.c_double),
        ("levels", ctypes.c_uint64),
    ]


def build(verbose=False):
    """Compile libsre.so for gfx950 (in-tree; the .so travels with the repo)."""
    cmd = ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-shared",
           "-fPIC", SRC, "-o", LIB]
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"hipcc failed:\n{r.stdout}\n{r.stderr}")
    if verbose:
        print(f"built {LIB}")


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(LIB):
            raise RuntimeError(
                f"{LIB} missing — run __graft_entry__.build() / reth_amd.engine.build()")
        _lib = ctypes.CDLL(LIB)
        _lib.sre_create.restype = ctypes.c_void_p
        _lib.sre_create.argtypes = [ctypes.c_int]
        _lib.sre_last_error.restype = ctypes.c_char_p
        _lib.sre_last_error.argtypes = [ctypes.c_void_p]
        _lib.sre_destroy.argtypes = [ctypes.c_void_p]
        _lib.sre_updates_count.restype = ctypes.c_int64
        _lib.sre_updates_count.argtypes = [ctypes.c_void_p]
    return _lib


def _np_ptr(arr):
    if len(arr) == 0:
        return None
    return arr.ctypes.data_as(ctypes.c_void_p)


class StateRootEngine:
    def __init__(self, device=0):
        self._lib = lib()
        self._ctx = self._lib.sre_create(device)
        if not self._ctx:
            raise RuntimeError("sre_create failed (GPU required, no fallback): "
                               + self._lib.sre_last_error(None).decode())
        self._keep = []  # borrowed device tensors kept alive

    def close(self):
        if getattr(self, "_ctx", None):
            self._lib.sre_destroy(self._ctx)
            self._ctx = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def _check(self, rc):
        if rc != 0:
            raise RuntimeError(self._lib.sre_last_error(
                ctypes.c_void_p(self._ctx)).decode())

    # ---- input upload ----
    def upload(self, accounts: np.ndarray, storage: np.ndarray):
        assert accounts.dtype == ACCOUNT_DTYPE and storage.dtype == STORAGE_DTYPE
        self._check(self._lib.sre_upload_accounts(
            ctypes.c_void_p(self._ctx), _np_ptr(accounts), len(accounts)))
        self._check(self._lib.sre_upload_storage(
            ctypes.c_void_p(self._ctx), _np_ptr(storage), len(storage)))

    def set_device_tensors(self, acct_u8, st_u8):
        """Borrow torch GPU tensors: acct (na,104) uint8, st (ns,96) uint8,
        contiguous, laid out as sre_account_entry / sre_storage_entry."""
        assert acct_u8.dtype.itemsize == 1 and acct_u8.is_contiguous()
        assert st_u8.dtype.itemsize == 1 and st_u8.is_contiguous()
        assert acct_u8.shape[1] == 104 and st_u8.shape[1] == 96
        self._keep = [acct_u8, st_u8]
        self._check(self._lib.sre_set_accounts_device(
            ctypes.c_void_p(self._ctx), ctypes.c_void_p(acct_u8.data_ptr()),
            acct_u8.shape[0]))
        self._check(self._lib.sre_set_storage_device(
            ctypes.c_void_p(self._ctx), ctypes.c_void_p(st_u8.data_ptr()),
            st_u8.shape[0]))

    def release_borrowed(self):
        """Drop the references keeping borrowed device tensors alive
        (set_device_tensors). Precondition: the engine no longer reads
        them — i.e. the resident state has been replaced by engine-owned
        arrays (apply_delta / incremental_root adopt merged copies first),
        or new tensors were set. After this the caller may free the
        tensors (torch.cuda.empty_cache())."""
        self._keep = []

    def apply_delta(self, acct_delta: np.ndarray, st_delta: np.ndarray):
        """Apply a HashedPostState overlay delta to the resident state
        (post-state wins, zero value deletes a slot, deleted accounts wipe
        their storage). A following root() computes the post-delta root —
        the incremental-root entry (BASELINE configs[4])."""
        assert acct_delta.dtype == DELTA_DTYPE
        assert st_delta.dtype == STORAGE_DTYPE
        self._check(self._lib.sre_apply_delta(
            ctypes.c_void_p(self._ctx), _np_ptr(acct_delta), len(acct_delta),
            _np_ptr(st_delta), len(st_delta)))

    # ---- compute ----
    def root_retaining(self) -> bytes:
        """Full root that also retains cell-top trie records and every
        account's storage root, so following incremental_root() calls
        recompute only what a delta touches. See include/sre.h
        sre_root_retaining."""
        out = (ctypes.c_uint8 * 32)()
        self._check(self._lib.sre_root_retaining(
            ctypes.c_void_p(self._ctx), out))
        return bytes(out)

    def incremental_root(self, acct_delta: np.ndarray,
                         st_delta: np.ndarray = None) -> bytes:
        """Apply a HashedPostState overlay delta (accounts + storage) and
        recompute the root along dirty paths only (requires a prior
        root_retaining): untouched accounts keep their retained storage
        roots, touched storage tries are rebuilt from their merged
        segments, and only delta-touched account-trie cells rehash. The
        resident state becomes the merged result and retention is
        refreshed, so deltas chain."""
        assert acct_delta.dtype == DELTA_DTYPE
        if st_delta is None:
            st_delta = np.zeros(0, dtype=STORAGE_DTYPE)
        assert st_delta.dtype == STORAGE_DTYPE
        out = (ctypes.c_uint8 * 32)()
        self._check(self._lib.sre_incremental_root(
            ctypes.c_void_p(self._ctx), _np_ptr(acct_delta),
            len(acct_delta), _np_ptr(st_delta), len(st_delta), out))
        return bytes(out)

    def root(self) -> bytes:
        out = (ctypes.c_uint8 * 32)()
        self._check(self._lib.sre_root(ctypes.c_void_p(self._ctx), out))
        return bytes(out)

    def _fetch_updates(self):
        n = self._lib.sre_updates_count(ctypes.c_void_p(self._ctx))
        rows = np.zeros(n, dtype=UPDATE_DTYPE)
        if n:
            self._check(self._lib.sre_updates_get(
                ctypes.c_void_p(self._ctx), _np_ptr(rows), n))
        return rows

    def root_retaining_with_updates(self):
        """Full root + full TrieUpdates row set, AND arms both the cell-top
        retention and the stored-row snapshot so following
        incremental_root_with_updates calls emit net row diffs
        (include/sre.h sre_root_retaining_with_updates)."""
        out = (ctypes.c_uint8 * 32)()
        self._check(self._lib.sre_root_retaining_with_updates(
            ctypes.c_void_p(self._ctx), out))
        return bytes(out), self._fetch_updates()

    def incremental_root_with_updates(self, acct_delta: np.ndarray,
                                      st_delta: np.ndarray = None):
        """Dirty-path incremental root + the NET TrieUpdates diff vs the
        pre-delta trie: upserts (removed=0), removals (removed=1,
        walker.rs:363-369 removed_nodes semantics) and destroyed-account
        whole-storage-trie markers (removed=2, updates.rs:154-157).
        Applying the diff to the pre-delta row set reproduces the full
        row set of the post-delta state. Requires a prior
        root_retaining_with_updates; deltas chain."""
        assert acct_delta.dtype == DELTA_DTYPE
        if st_delta is None:
            st_delta = np.zeros(0, dtype=STORAGE_DTYPE)
        assert st_delta.dtype == STORAGE_DTYPE
        out = (ctypes.c_uint8 * 32)()
        self._check(self._lib.sre_incremental_root_with_updates(
            ctypes.c_void_p(self._ctx), _np_ptr(acct_delta),
            ctypes.c_uint64(len(acct_delta)), _np_ptr(st_delta),
            ctypes.c_uint64(len(st_delta)), out))
        return bytes(out), self._fetch_updates()

    def root_from_nodes(self, rows: np.ndarray, acct_delta: np.ndarray = None,
                        st_delta: np.ndarray = None) -> bytes:
        """state_root_from_nodes / TrieInput equivalent: root of
        (resident state + delta) with `rows` (UPDATE_DTYPE, e.g. from
        root_with_updates) seeding untouched accounts' storage roots so
        their tries are not recomputed (include/sre.h
        sre_root_from_nodes). Replaces the resident state with the
        merged result."""
        assert rows.dtype == UPDATE_DTYPE
        if acct_delta is None:
            acct_delta = np.zeros(0, dtype=DELTA_DTYPE)
        if st_delta is None:
            st_delta = np.zeros(0, dtype=STORAGE_DTYPE)
        assert acct_delta.dtype == DELTA_DTYPE
        assert st_delta.dtype == STORAGE_DTYPE
        out = (ctypes.c_uint8 * 32)()
        # 8 args: stack-passed lengths MUST be explicit c_uint64 (a bare
        # int becomes a 32-bit stack slot with a garbage upper half)
        self._check(self._lib.sre_root_from_nodes(
            ctypes.c_void_p(self._ctx), _np_ptr(rows),
            ctypes.c_uint64(len(rows)),
            _np_ptr(acct_delta), ctypes.c_uint64(len(acct_delta)),
            _np_ptr(st_delta), ctypes.c_uint64(len(st_delta)), out))
        return bytes(out)

    def root_with_updates(self):
        """State root + TrieUpdates rows (UPDATE_DTYPE), sorted like reth's
        TrieUpdates::into_sorted. Surface of
        StateRootProvider::state_root_with_updates."""
        out = (ctypes.c_uint8 * 32)()
        self._check(self._lib.sre_root_with_updates(
            ctypes.c_void_p(self._ctx), out))
        return bytes(out), self._fetch_updates()

    def account_proof(self, targets) -> list:
        """Account multiproof: per target, the proof node RLPs root-first
        (Proof::multiproof account surface, include/sre.h
        sre_account_proof). Present keys get the path to their leaf;
        absent keys get the exclusion proof ending at the divergence.
        targets: list of 32-byte hashed keys."""
        n = len(targets)
        tarr = np.frombuffer(b"".join(targets), dtype=np.uint8).reshape(n, 32)
        tarr = np.ascontiguousarray(tarr)
        cap_nodes = n * 130 * 560 + 4096
        cap_lens = n * 132
        nodes = np.zeros(cap_nodes, dtype=np.uint8)
        lens = np.zeros(cap_lens, dtype=np.uint32)
        counts = np.zeros(n, dtype=np.uint32)
        self._check(self._lib.sre_account_proof(
            ctypes.c_void_p(self._ctx), _np_ptr(tarr), ctypes.c_uint64(n),
            _np_ptr(nodes), ctypes.c_uint64(cap_nodes),
            _np_ptr(lens), ctypes.c_uint64(cap_lens), _np_ptr(counts)))
        out, off, li = [], 0, 0
        for t in range(n):
            tl = []
            for _ in range(int(counts[t])):
                ln = int(lens[li]); li += 1
                tl.append(nodes[off:off + ln].tobytes()); off += ln
            out.append(tl)
        return out

    def storage_proof(self, acct_keys, slot_keys):
        """Storage multiproof for (account, slot) pairs. The slot may be
        absent (exclusion proof); an absent or storage-less account
        yields (EMPTY_ROOT_HASH, []) = StorageMultiProof::empty().
        Returns (roots, proofs) — per target the account's storage root
        and the root-first node-RLP list of its storage trie."""
        n = len(acct_keys)
        assert len(slot_keys) == n
        aarr = np.ascontiguousarray(
            np.frombuffer(b"".join(acct_keys), dtype=np.uint8).reshape(n, 32))
        sarr = np.ascontiguousarray(
            np.frombuffer(b"".join(slot_keys), dtype=np.uint8).reshape(n, 32))
        cap_nodes = n * 130 * 560 + 4096
        cap_lens = n * 132
        roots = np.zeros((n, 32), dtype=np.uint8)
        nodes = np.zeros(cap_nodes, dtype=np.uint8)
        lens = np.zeros(cap_lens, dtype=np.uint32)
        counts = np.zeros(n, dtype=np.uint32)
        self._check(self._lib.sre_storage_proof(
            ctypes.c_void_p(self._ctx), _np_ptr(aarr), _np_ptr(sarr),
            ctypes.c_uint64(n), _np_ptr(roots),
            _np_ptr(nodes), ctypes.c_uint64(cap_nodes),
            _np_ptr(lens), ctypes.c_uint64(cap_lens), _np_ptr(counts)))
        out, off, li = [], 0, 0
        for t in range(n):
            tl = []
            for _ in range(int(counts[t])):
                ln = int(lens[li]); li += 1
                tl.append(nodes[off:off + ln].tobytes()); off += ln
            out.append(tl)
        return [bytes(r) for r in roots], out

    def storage_roots(self, n) -> np.ndarray:
        out = np.empty((n, 32), dtype=np.uint8)
        self._check(self._lib.sre_storage_roots(
            ctypes.c_void_p(self._ctx), _np_ptr(out), n))
        return out

    def subtree_roots(self):
        refs = np.zeros((16, 33), dtype=np.uint8)
        lens = np.zeros(16, dtype=np.uint8)
        roots = np.zeros((16, 32), dtype=np.uint8)
        counts = np.zeros(16, dtype=np.uint64)
        self._check(self._lib.sre_subtree_roots(
            ctypes.c_void_p(self._ctx), _np_ptr(refs), _np_ptr(lens),
            _np_ptr(roots), _np_ptr(counts)))
        return refs, lens, roots, counts

    def finish_top(self, refs, lens, roots, counts) -> bytes:
        out = (ctypes.c_uint8 * 32)()
        self._check(self._lib.sre_finish_top(
            ctypes.c_void_p(self._ctx),
            _np_ptr(np.ascontiguousarray(refs, dtype=np.uint8)),
            _np_ptr(np.ascontiguousarray(lens, dtype=np.uint8)),
            _np_ptr(np.ascontiguousarray(roots, dtype=np.uint8)),
            _np_ptr(np.ascontiguousarray(counts, dtype=np.uint64)), out))
        return bytes(out)

    def keccak_batch_device(self, in_tensor, msg_len, out_tensor):
        """Hash n messages of msg_len bytes at a fixed stride (device memory).
        in_tensor: (n, stride) uint8 cuda tensor; out_tensor: (n, 32) uint8.

        The kernel runs on the engine's own HIP stream: drain torch's stream
        first so pending writes to in_tensor are visible (the engine
        synchronizes its stream before returning, so later torch ops are
        safe the other way around)."""
        import torch
        if in_tensor.is_cuda:
            torch.cuda.synchronize(in_tensor.device)
        n, stride = in_tensor.shape
        self._check(self._lib.sre_keccak_batch_device(
            ctypes.c_void_p(self._ctx), ctypes.c_void_p(in_tensor.data_ptr()),
            ctypes.c_uint64(stride), ctypes.c_uint32(msg_len), ctypes.c_uint64(n),
            ctypes.c_void_p(out_tensor.data_ptr())))
        return out_tensor

    def stats(self) -> dict:
        s = SreStats()
        self._check(self._lib.sre_get_stats(ctypes.c_void_p(self._ctx),
                                            ctypes.byref(s)))
        return {f[0]: getattr(s, f[0]) for f in SreStats._fields_}
