"""Parity of the closed-form small-delta merges with the gate lowered so
the path runs on test-sized states (SRE_ST_SMALL_MIN=0): the same chained
storage/destruction scenarios as the general path, checked against the
oracle on dict-merged state. Covers wipe ranges (destroyed accounts with
storage), slot deletes/upserts, inserts at segment boundaries, and the
keep-storage shortcut for accounts-only deltas over storage state."""
import os

import numpy as np
import pytest

from oracle import bind
from reth_amd import gen
from reth_amd.engine import DELTA_DTYPE
from tests.test_gpu_incremental import (_dict_of, _arrays_of, _apply_dict,
                                        _mk_delta)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    os.environ["SRE_ST_SMALL_MIN"] = "0"  # getenv is read per call
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()
    os.environ.pop("SRE_ST_SMALL_MIN", None)


def test_smallmerge_storage_chained(eng):
    rng = np.random.default_rng(2718)
    acct, st = gen.gen_state_numpy(3000, 6, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    assert eng.root() == bind.state_root(*_arrays_of(accounts))
    ke = bind.keccak256(b"")
    for step in range(5):
        keys = sorted(accounts)
        rows, strows = [], []
        # slot upserts + deletions
        for i in rng.choice(len(keys), 15, replace=False):
            k = keys[int(i)]
            slots = accounts[k][3]
            if slots and rng.random() < 0.4:
                dead = sorted(slots)[0]
                strows.append((k, dead, 0))
                del slots[dead]
            nk = bind.keccak256(b"sm" + bytes([step]) + k[:4])
            strows.append((k, nk, 4000 + step))
            slots[nk] = 4000 + step
        # destroy two storage-bearing accounts (wipe ranges)
        nburn = 0
        for k in list(keys):
            if nburn >= 2:
                break
            if accounts[k][3] and all(r[0] != k for r in strows):
                rows.append((k, 0, 0, ke, 1))
                del accounts[k]
                nburn += 1
        # insert an account with storage (insert at a fresh segment)
        nk = bind.keccak256(b"sm-acct" + bytes([step]))
        ns1 = bind.keccak256(b"sm-slot" + bytes([step]))
        rows.append((nk, 1, 3, ke, 0))
        strows.append((nk, ns1, 55))
        accounts[nk] = [1, 3, ke, {ns1: 55}]
        rows = sorted(set(rows))
        strows = sorted(set(strows))
        d, s = _mk_delta(rows, strows)
        eng.apply_delta(d, s)
        assert eng.root() == bind.state_root(*_arrays_of(accounts)), \
            f"step {step}"


def test_smallmerge_keep_storage_accounts_only(eng):
    # accounts-only delta over engine-OWNED storage: the keep-in-place
    # shortcut must leave the storage intact
    acct, st = gen.gen_state_numpy(800, 4, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)  # upload copies -> engine-owned
    ke = bind.keccak256(b"")
    keys = sorted(accounts)
    rows = []
    for k in keys[5:25]:
        accounts[k][1] += 77
        rows.append((k, accounts[k][0], accounts[k][1], accounts[k][2], 0))
    nk = bind.keccak256(b"km-new")
    rows.append((nk, 1, 9, ke, 0))
    accounts[nk] = [1, 9, ke, {}]
    d, s = _mk_delta(sorted(rows), [])
    eng.apply_delta(d, s)
    assert eng.root() == bind.state_root(*_arrays_of(accounts))


def test_smallmerge_dirty_path_with_storage(eng):
    # the closed-form storage merge under the dirty-path incremental
    rng = np.random.default_rng(99)
    acct, st = gen.gen_state_numpy(2000, 4, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    assert eng.root_retaining() == bind.state_root(*_arrays_of(accounts))
    ke = bind.keccak256(b"")
    for step in range(4):
        keys = sorted(accounts)
        rows, strows = [], []
        for i in rng.choice(len(keys), 8, replace=False):
            k = keys[int(i)]
            nk = bind.keccak256(b"dp" + bytes([step]) + k[:3])
            strows.append((k, nk, 123 + step))
            accounts[k][3][nk] = 123 + step
        victim = next(k for k in keys if accounts[k][3]
                      and all(r[0] != k for r in strows))
        rows.append((victim, 0, 0, ke, 1))
        del accounts[victim]
        rows = sorted(set(rows))
        strows = sorted(set(strows))
        d, s = _mk_delta(rows, strows)
        assert eng.incremental_root(d, s) == \
            bind.state_root(*_arrays_of(accounts)), f"step {step}"


@pytest.fixture()
def force_small():
    # SRE_SD_FORCE lowers the account small-path gate (nd <= na instead
    # of nd <= na/64) so test-sized deltas take apply_delta_small
    os.environ["SRE_SD_FORCE"] = "1"
    yield
    os.environ.pop("SRE_SD_FORCE", None)


def test_smallmerge_forced_account_path_chained(eng, force_small):
    rng = np.random.default_rng(777)
    acct, _ = gen.gen_state_numpy(1500, 0, bind.keccak256_batch)
    accounts = {bytes(a["key"]): [int(a["nonce"]),
                int.from_bytes(bytes(a["balance"]), "big"),
                bytes(a["code_hash"]), {}] for a in acct}
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    assert eng.root_retaining() == bind.state_root(*_arrays_of(accounts))
    ke = bind.keccak256(b"")
    for step in range(4):
        keys = sorted(accounts)
        rows = []
        for i in rng.choice(len(keys), 60, replace=False):
            k = keys[int(i)]
            v = accounts[k]
            rows.append((k, v[0] + 1, v[1] + 3, v[2], 0))
        for i in rng.choice(len(keys), 20, replace=False):
            k = keys[int(i)]
            if any(r[0] == k for r in rows):
                continue
            rows.append((k, 0, 0, ke, 1))
        for i in range(25):
            rows.append((bind.keccak256(b"fs" + bytes([step, i])),
                         1, 100 + i, ke, 0))
        rows = sorted(set(rows))
        _apply_dict(accounts, rows)
        d, _ = _mk_delta(rows, [])
        assert eng.incremental_root(d) == \
            bind.state_root(*_arrays_of(accounts)), f"step {step}"


def test_smallmerge_forced_with_updates(eng, force_small):
    # the small merge + incremental TrieUpdates diff combination
    from tests.test_gpu_incremental_updates import (_rowmap, _apply_diff)
    rng = np.random.default_rng(31415)
    acct, _ = gen.gen_state_numpy(2000, 0, bind.keccak256_batch)
    accounts = {bytes(a["key"]): [int(a["nonce"]),
                int.from_bytes(bytes(a["balance"]), "big"),
                bytes(a["code_hash"]), {}] for a in acct}
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    root0, rows0 = eng.root_retaining_with_updates()
    cur = _rowmap(rows0)
    ke = bind.keccak256(b"")
    for step in range(3):
        keys = sorted(accounts)
        rows = []
        for i in rng.choice(len(keys), 50, replace=False):
            k = keys[int(i)]
            v = accounts[k]
            rows.append((k, v[0] + 2, v[1] + 9, v[2], 0))
        for i in rng.choice(len(keys), 15, replace=False):
            k = keys[int(i)]
            if any(r[0] == k for r in rows):
                continue
            rows.append((k, 0, 0, ke, 1))
        rows.append((bind.keccak256(b"fw" + bytes([step])), 1, 7, ke, 0))
        rows = sorted(set(rows))
        _apply_dict(accounts, rows)
        d, _ = _mk_delta(rows, [])
        root, diff = eng.incremental_root_with_updates(d)
        oroot, orows = bind.state_root_with_updates(*_arrays_of(accounts))
        assert root == oroot, f"step {step}"
        cur = _apply_diff(cur, diff)
        assert cur == _rowmap(orows), f"step {step}"


def test_smallmerge_forced_delete_all(eng, force_small):
    ke = bind.keccak256(b"")
    acct, _ = gen.gen_state_numpy(300, 0, bind.keccak256_batch)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    eng.root_retaining()
    # one small delta first (arms the lcp repair chain), then delete all
    keys = sorted(bytes(a["key"]) for a in acct)
    d1, _ = _mk_delta([(keys[0], 9, 9, ke, 0)], [])
    eng.incremental_root(d1)
    rows = [(k, 0, 0, ke, 1) for k in keys]
    d, _ = _mk_delta(sorted(rows), [])
    empty_root = bytes.fromhex(
        "56e81f171bcc55a6ff8345e692c0f86e5b48e01b996cadc001622fb5e363b421")
    assert eng.incremental_root(d) == empty_root


def test_smallmerge_storage_with_updates_diff(eng):
    # closed-form STORAGE merge (gate lowered module-wide) under the
    # incremental TrieUpdates diff: wipe ranges + slot churn must yield a
    # net diff that reproduces the oracle's full-rebuild row set
    from tests.test_gpu_incremental_updates import (_rowmap, _apply_diff)
    rng = np.random.default_rng(60221023)
    acct, st = gen.gen_state_numpy(1200, 5, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    root0, rows0 = eng.root_retaining_with_updates()
    oroot, orows = bind.state_root_with_updates(*_arrays_of(accounts))
    assert root0 == oroot
    cur = _rowmap(rows0)
    assert cur == _rowmap(orows)
    ke = bind.keccak256(b"")
    for step in range(3):
        keys = sorted(accounts)
        rows, strows = [], []
        for i in rng.choice(len(keys), 10, replace=False):
            k = keys[int(i)]
            slots = accounts[k][3]
            if slots and rng.random() < 0.5:
                dead = sorted(slots)[0]
                strows.append((k, dead, 0))
                del slots[dead]
            nk = bind.keccak256(b"su" + bytes([step]) + k[:4])
            strows.append((k, nk, 777 + step))
            slots[nk] = 777 + step
        victim = next(k for k in keys if accounts[k][3]
                      and all(r[0] != k for r in strows))
        rows.append((victim, 0, 0, ke, 1))
        del accounts[victim]
        rows = sorted(set(rows))
        strows = sorted(set(strows))
        d, s = _mk_delta(rows, strows)
        root, diff = eng.incremental_root_with_updates(d, s)
        oroot, orows = bind.state_root_with_updates(*_arrays_of(accounts))
        assert root == oroot, f"step {step}"
        assert bytes(victim) in {bytes(r["acct_key"])
                                 for r in diff[diff["removed"] == 2]}
        cur = _apply_diff(cur, diff)
        assert cur == _rowmap(orows), f"step {step}"


def test_smallmerge_rejects_unsorted_account_delta(eng, force_small):
    acct, _ = gen.gen_state_numpy(200, 0, bind.keccak256_batch)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    eng.root_retaining()
    ke = bind.keccak256(b"")
    # warm delta arms the retained lcp so the NEXT call takes the small path
    warm, _ = _mk_delta([(bytes(acct[0]["key"]), 5, 5, ke, 0)], [])
    eng.incremental_root(warm)
    k1, k2 = sorted([bind.keccak256(b"u1"), bind.keccak256(b"u2")])
    d = np.zeros(2, DELTA_DTYPE)
    for i, k in enumerate([k2, k1]):  # wrong order
        d[i]["key"] = np.frombuffer(k, np.uint8)
        d[i]["nonce"] = 1
        d[i]["code_hash"] = np.frombuffer(ke, np.uint8)
    with pytest.raises(RuntimeError):
        eng.incremental_root(d)


def test_smallmerge_rejects_bad_delta(eng):
    acct, st = gen.gen_state_numpy(200, 2, bind.keccak256_batch)
    eng.upload(acct, st)
    k = bytes(acct[0]["key"])
    # storage row for a destroyed account must be rejected on this path too
    d = np.zeros(1, DELTA_DTYPE)
    d[0]["key"] = np.frombuffer(k, np.uint8)
    d[0]["deleted"] = 1
    s = np.zeros(1, bind.STORAGE_DTYPE)
    s[0]["acct_key"] = np.frombuffer(k, np.uint8)
    s[0]["slot_key"] = np.frombuffer(bind.keccak256(b"x"), np.uint8)
    s[0]["value"] = np.frombuffer((1).to_bytes(32, "big"), np.uint8)
    with pytest.raises(RuntimeError):
        eng.apply_delta(d, s)
