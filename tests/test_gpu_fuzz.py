"""Randomized end-to-end parity fuzz: many small random states in one GPU
session, each checked root / storage-roots / TrieUpdates / proofs against
the CPU oracle. Catches cross-feature interactions the targeted suites
miss (buffer reuse across calls, retention invalidation, proof capture on
reused contexts)."""
import numpy as np
import pytest

from oracle import bind, pyref
from reth_amd.engine import DELTA_DTYPE
from tests.util import to_arrays
from tests.test_gpu_proof import _dict_of, _replay

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()


def _rand_state(rng, na, slots):
    accounts = {}
    for _ in range(na):
        k = bind.keccak256(rng.bytes(8))
        sl = {bind.keccak256(rng.bytes(8)): int(rng.integers(1, 2**63))
              for _ in range(slots)}
        accounts[k] = (int(rng.integers(0, 2**32)),
                       int(rng.integers(0, 2**62)) << int(rng.integers(0, 40)),
                       bind.keccak256(rng.bytes(4)), sl)
    return accounts


FUZZ_ITERS = int(__import__("os").environ.get("SRE_FUZZ_ITERS", "12"))


def test_fuzz_roots_updates_proofs(eng):
    rng = np.random.default_rng(0xF00D)
    for it in range(FUZZ_ITERS):
        na = int(rng.integers(1, 400))
        slots = int(rng.integers(0, 6))
        accounts = _rand_state(rng, na, slots)
        acct, st = to_arrays(accounts)
        eng.upload(acct, st)
        root = bind.state_root(acct, st)
        assert eng.root() == root, f"iter {it}: root"
        # updates parity on a third of the iterations (row-level compare)
        if it % 3 == 0:
            r2, rows = eng.root_with_updates()
            assert r2 == root
            oroot, orows = bind.state_root_with_updates(acct, st)
            assert oroot == root
            assert len(rows) == len(orows), f"iter {it}: update row count"
            for a, b in zip(rows, orows):
                for f in ("kind", "path_len", "state_mask", "tree_mask",
                          "hash_mask", "num_hashes", "root_hash_set"):
                    assert a[f] == b[f], f"iter {it}: {f}"
                assert bytes(a["path"]) == bytes(b["path"])
                assert bytes(a["hashes"].tobytes()) == bytes(b["hashes"].tobytes())
        # proofs for a few random present accounts
        keys = sorted(accounts)
        tsel = [keys[int(i)] for i in
                rng.choice(len(keys), min(3, len(keys)), replace=False)]
        tsel += [bind.keccak256(b"absent" + bytes([it, j])) for j in range(2)]
        proofs = eng.account_proof(tsel)
        for k, nodes in zip(tsel, proofs):
            assert nodes == pyref.account_proof(accounts, k), f"iter {it}"
            got = _replay(nodes, k, root)
            assert (got is None) == (k not in accounts), f"iter {it}"


def test_fuzz_incremental_chains(eng):
    rng = np.random.default_rng(0xBEEF)
    ke = bind.keccak256(b"")
    for it in range(max(5, FUZZ_ITERS // 2)):
        na = int(rng.integers(2, 300))
        accounts = {k: list(v[:3]) + [{}]
                    for k, v in _rand_state(rng, na, 0).items()}
        acct, _ = to_arrays({k: tuple(v) for k, v in accounts.items()})
        eng.upload(acct, np.zeros(0, dtype=bind.STORAGE_DTYPE))
        assert eng.root_retaining() == bind.state_root(
            *to_arrays({k: tuple(v) for k, v in accounts.items()}))
        for step in range(3):
            keys = sorted(accounts)
            rows = []
            for i in rng.choice(len(keys), min(8, len(keys)), replace=False):
                k = keys[int(i)]
                if rng.random() < 0.3:
                    rows.append((k, 0, 0, ke, 1))
                    del accounts[k]
                else:
                    accounts[k][1] += 1
                    rows.append((k, accounts[k][0], accounts[k][1],
                                 accounts[k][2], 0))
            nk = bind.keccak256(b"fz" + bytes([it, step]))
            accounts[nk] = [1, 7, ke, {}]
            rows.append((nk, 1, 7, ke, 0))
            rows = sorted(set(rows))
            d = np.zeros(len(rows), dtype=DELTA_DTYPE)
            for i, (k, n, b, ch, dead) in enumerate(rows):
                d[i]["key"] = np.frombuffer(k, np.uint8)
                d[i]["nonce"] = n
                d[i]["balance"] = np.frombuffer(b.to_bytes(32, "big"), np.uint8)
                d[i]["code_hash"] = np.frombuffer(ch, np.uint8)
                d[i]["deleted"] = dead
            want = bind.state_root(
                *to_arrays({k: tuple(v) for k, v in accounts.items()}))
            assert eng.incremental_root(d) == want, f"iter {it} step {step}"
