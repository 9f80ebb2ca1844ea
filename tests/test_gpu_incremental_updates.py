"""Incremental TrieUpdates parity: sre_incremental_root_with_updates must
emit the NET stored-row diff of each delta — upserts, removed paths
(walker.rs:363-369 removed_nodes semantics) and destroyed-account
whole-storage-trie markers (updates.rs:154-157) — such that applying the
diff to the pre-delta row set reproduces the full-rebuild row set of the
post-delta state (the oracle's dict-rebuild update diff)."""
import numpy as np
import pytest

from oracle import bind
from reth_amd import gen
from reth_amd.engine import DELTA_DTYPE
from tests.test_gpu_incremental import (_dict_of, _arrays_of, _acct_dict,
                                        _apply_dict, _mk_delta)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()


def _key(r):
    pl = int(r["path_len"])
    return (int(r["kind"]), bytes(r["acct_key"]), pl,
            bytes(r["path"])[:(pl + 1) // 2])


def _content(r):
    nh = int(r["num_hashes"])
    return (int(r["state_mask"]), int(r["tree_mask"]), int(r["hash_mask"]),
            nh, bytes(r["hashes"][:nh].tobytes()),
            int(r["root_hash_set"]), bytes(r["root_hash"]))


def _rowmap(rows):
    m = {}
    for r in rows:
        k = _key(r)
        assert k not in m, "duplicate row"
        m[k] = _content(r)
    return m


def _oracle_rows(accounts):
    root, rows = bind.state_root_with_updates(*_arrays_of(accounts))
    return root, _rowmap(rows)


def _apply_diff(rows_map, diff):
    """Apply the engine's net diff to a row map, checking its claims."""
    for r in diff[diff["removed"] == 2]:
        assert int(r["kind"]) == 1 and int(r["path_len"]) == 0
        ak = bytes(r["acct_key"])
        for k in [k for k in rows_map if k[0] == 1 and k[1] == ak]:
            del rows_map[k]
    for r in diff:
        rem = int(r["removed"])
        if rem == 2:
            continue
        k = _key(r)
        if rem == 1:
            assert k in rows_map, f"removal of unknown row {k}"
            del rows_map[k]
        else:
            # net diff: upserts must actually be new or changed
            assert rows_map.get(k) != _content(r), f"unchanged re-emit {k}"
            rows_map[k] = _content(r)
    return rows_map


def test_incremental_updates_accounts_only_chained(eng):
    rng = np.random.default_rng(4321)
    acct, _ = gen.gen_state_numpy(5000, 0, bind.keccak256_batch)
    accounts = _acct_dict(acct)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    root0, rows0 = eng.root_retaining_with_updates()
    oroot, omap = _oracle_rows(accounts)
    assert root0 == oroot
    cur = _rowmap(rows0)
    assert cur == omap
    ke = bind.keccak256(b"")
    for step in range(5):
        keys = sorted(accounts)
        rows = []
        for i in rng.choice(len(keys), size=60, replace=False):
            k = keys[int(i)]
            v = accounts[k]
            rows.append((k, v[0] + 1, v[1] + 3, v[2], 0))
        for i in rng.choice(len(keys), size=20, replace=False):
            k = keys[int(i)]
            if any(r[0] == k for r in rows):
                continue
            rows.append((k, 0, 0, ke, 1))
        for i in range(25):
            nk = bind.keccak256(b"iu" + bytes([step, i]))
            rows.append((nk, 2, 500 + i, ke, 0))
        rows = sorted(set(rows))
        _apply_dict(accounts, rows)
        d, _ = _mk_delta(rows, [])
        root, diff = eng.incremental_root_with_updates(d)
        oroot, omap = _oracle_rows(accounts)
        assert root == oroot, f"step {step}"
        cur = _apply_diff(cur, diff)
        assert cur == omap, f"step {step}: diff does not reproduce rows"


def test_incremental_updates_with_storage_and_destruction(eng):
    rng = np.random.default_rng(555)
    acct, st = gen.gen_state_numpy(1500, 5, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    root0, rows0 = eng.root_retaining_with_updates()
    oroot, omap = _oracle_rows(accounts)
    assert root0 == oroot
    cur = _rowmap(rows0)
    assert cur == omap
    ke = bind.keccak256(b"")
    for step in range(4):
        keys = sorted(accounts)
        rows, strows = [], []
        # storage churn on random accounts
        for i in rng.choice(len(keys), 10, replace=False):
            k = keys[int(i)]
            slots = accounts[k][3]
            if slots and rng.random() < 0.5:
                dead = sorted(slots)[0]
                strows.append((k, dead, 0))
                del slots[dead]
            nk = bind.keccak256(b"us" + bytes([step]) + k[:4])
            strows.append((k, nk, 31337 + step))
            slots[nk] = 31337 + step
        # destroy one storage-bearing account -> removed=2 marker expected
        victim = next(k for k in keys if accounts[k][3]
                      and all(r[0] != k for r in strows))
        rows.append((victim, 0, 0, ke, 1))
        del accounts[victim]
        # new account with storage
        nk = bind.keccak256(b"ui-acct" + bytes([step]))
        ns = bind.keccak256(b"ui-slot" + bytes([step]))
        rows.append((nk, 1, 7, ke, 0))
        strows.append((nk, ns, 99))
        accounts[nk] = [1, 7, ke, {ns: 99}]
        rows = sorted(set(rows))
        strows = sorted(set(strows))
        d, s = _mk_delta(rows, strows)
        root, diff = eng.incremental_root_with_updates(d, s)
        oroot, omap = _oracle_rows(accounts)
        assert root == oroot, f"step {step}"
        markers = diff[diff["removed"] == 2]
        assert bytes(victim) in {bytes(r["acct_key"]) for r in markers}, \
            "destroyed account must get a whole-trie deletion marker"
        cur = _apply_diff(cur, diff)
        assert cur == omap, f"step {step}: diff does not reproduce rows"


def test_incremental_updates_suppresses_unchanged(eng):
    acct, _ = gen.gen_state_numpy(800, 0, bind.keccak256_batch)
    accounts = _acct_dict(acct)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    root0, _ = eng.root_retaining_with_updates()
    # empty delta -> empty diff
    root, diff = eng.incremental_root_with_updates(np.zeros(0, DELTA_DTYPE))
    assert root == root0 and len(diff) == 0
    # idempotent delta (rewrite identical values) -> empty diff
    keys = sorted(accounts)
    rows = [(k, accounts[k][0], accounts[k][1], accounts[k][2], 0)
            for k in keys[:30]]
    d, _ = _mk_delta(rows, [])
    root, diff = eng.incremental_root_with_updates(d)
    assert root == root0 and len(diff) == 0


def test_incremental_updates_delete_everything(eng):
    # a delta that empties the state: every stored row becomes a removal
    # (plus whole-trie markers for storage-bearing destroyed accounts)
    acct, st = gen.gen_state_numpy(400, 3, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    root0, rows0 = eng.root_retaining_with_updates()
    cur = _rowmap(rows0)
    ke = bind.keccak256(b"")
    rows = [(k, 0, 0, ke, 1) for k in sorted(accounts)]
    d, _ = _mk_delta(rows, [])
    root, diff = eng.incremental_root_with_updates(d)
    empty_root = bytes.fromhex(
        "56e81f171bcc55a6ff8345e692c0f86e5b48e01b996cadc001622fb5e363b421")
    assert root == empty_root
    cur = _apply_diff(cur, diff)
    assert cur == {}, "all rows must be removed"
    assert not any(int(r["removed"]) == 0 for r in diff), "no upserts"


def test_incremental_updates_requires_arming(eng):
    acct, _ = gen.gen_state_numpy(100, 0, bind.keccak256_batch)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    eng.root_retaining()  # plain retention does NOT arm the row snapshot
    with pytest.raises(RuntimeError):
        eng.incremental_root_with_updates(np.zeros(0, DELTA_DTYPE))


def test_incremental_updates_ordering(eng):
    # rows come out like TrieUpdatesSorted: account rows first (nibble-path
    # sorted, upserts and removals interleaved), then storage rows grouped
    # by account
    rng = np.random.default_rng(8)
    acct, st = gen.gen_state_numpy(1200, 4, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    eng.root_retaining_with_updates()
    ke = bind.keccak256(b"")
    keys = sorted(accounts)
    rows, strows = [], []
    for i in rng.choice(len(keys), 30, replace=False):
        k = keys[int(i)]
        rows.append((k, 9, 9, accounts[k][2], 0))
        accounts[k][0] = 9
        accounts[k][1] = 9
        nk = bind.keccak256(b"ord" + k[:4])
        strows.append((k, nk, 5))
        accounts[k][3][nk] = 5
    d, s = _mk_delta(sorted(rows), sorted(strows))
    root, diff = eng.incremental_root_with_updates(d, s)
    oroot, omap = _oracle_rows(accounts)
    assert root == oroot
    sortkeys = []
    for r in diff:
        pl = int(r["path_len"])
        nibs = []
        pb = bytes(r["path"])
        for k in range(pl):
            nibs.append((pb[k // 2] >> (0 if k % 2 else 4)) & 0xF)
        sortkeys.append((int(r["kind"]), bytes(r["acct_key"]), nibs))
    assert sortkeys == sorted(sortkeys)
