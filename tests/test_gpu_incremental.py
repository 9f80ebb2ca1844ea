"""Overlay-delta (incremental root) parity: sre_apply_delta must reproduce
the HashedPostState overlay semantics (post-state wins, zero deletes,
destroyed accounts wipe storage — post_state.rs:89,313,355), checked by
applying the same delta to a python dict and comparing roots via the CPU
oracle."""
import numpy as np
import pytest

from oracle import bind
from reth_amd import gen
from reth_amd.engine import DELTA_DTYPE
from tests.util import to_arrays

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()


def _dict_of(acct, st):
    accounts = {}
    for a in acct:
        accounts[bytes(a["key"])] = [int(a["nonce"]),
                                     int.from_bytes(bytes(a["balance"]), "big"),
                                     bytes(a["code_hash"]), {}]
    for s in st:
        accounts[bytes(s["acct_key"])][3][bytes(s["slot_key"])] = \
            int.from_bytes(bytes(s["value"]), "big")
    return accounts


def _arrays_of(accounts):
    return to_arrays({k: tuple(v) for k, v in accounts.items()})


def _mk_delta(rows, strows):
    """rows: [(key, nonce, balance, code_hash, deleted)]; strows:
    [(acct_key, slot_key, value_int)] — zero value = delete."""
    d = np.zeros(len(rows), dtype=DELTA_DTYPE)
    for i, (k, n, b, ch, dead) in enumerate(sorted(rows)):
        d[i]["key"] = np.frombuffer(k, np.uint8)
        d[i]["nonce"] = n
        d[i]["balance"] = np.frombuffer(b.to_bytes(32, "big"), np.uint8)
        d[i]["code_hash"] = np.frombuffer(ch, np.uint8)
        d[i]["deleted"] = dead
    s = np.zeros(len(strows), dtype=bind.STORAGE_DTYPE)
    for i, (a, sk, v) in enumerate(sorted(strows)):
        s[i]["acct_key"] = np.frombuffer(a, np.uint8)
        s[i]["slot_key"] = np.frombuffer(sk, np.uint8)
        s[i]["value"] = np.frombuffer(v.to_bytes(32, "big"), np.uint8)
    return d, s


def test_overlay_delta_parity(eng):
    acct, st = gen.gen_state_numpy(3000, 4, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    keys = sorted(accounts)
    ke = bind.keccak256(b"")

    rows, strows = [], []
    # modify 10 accounts' balances
    for k in keys[10:20]:
        v = accounts[k]
        v[1] += 12345
        rows.append((k, v[0], v[1], v[2], 0))
    # delete 5 accounts (their storage must vanish)
    for k in keys[30:35]:
        rows.append((k, 0, 0, ke, 1))
        del accounts[k]
    # insert 7 new accounts, two with fresh storage slots
    for i in range(7):
        nk = bind.keccak256(b"new" + bytes([i]))
        slots = {}
        if i < 2:
            sk = bind.keccak256(b"ns" + bytes([i]))
            slots[sk] = 777 + i
            strows.append((nk, sk, 777 + i))
        accounts[nk] = [5, 999 + i, ke, slots]
        rows.append((nk, 5, 999 + i, ke, 0))
    # slot upserts + slot deletions on a surviving account
    tgt = keys[50]
    tslots = accounts[tgt][3]
    if tslots:
        dead_slot = sorted(tslots)[0]
        strows.append((tgt, dead_slot, 0))  # delete
        del tslots[dead_slot]
    newslot = bind.keccak256(b"fresh")
    strows.append((tgt, newslot, 42))
    tslots[newslot] = 42

    want = bind.state_root(*_arrays_of(accounts))

    eng.upload(acct, st)
    d, s = _mk_delta(rows, strows)
    eng.apply_delta(d, s)
    assert eng.root() == want
    # empty delta: unchanged
    eng.apply_delta(np.zeros(0, DELTA_DTYPE), np.zeros(0, bind.STORAGE_DTYPE))
    assert eng.root() == want


def test_overlay_on_empty_base(eng):
    eng.upload(np.zeros(0, bind.ACCOUNT_DTYPE), np.zeros(0, bind.STORAGE_DTYPE))
    ke = bind.keccak256(b"")
    k1, k2 = sorted([bind.keccak256(b"a"), bind.keccak256(b"b")])
    sk = bind.keccak256(b"s")
    d, s = np.zeros(2, DELTA_DTYPE), np.zeros(1, bind.STORAGE_DTYPE)
    for i, k in enumerate([k1, k2]):
        d[i]["key"] = np.frombuffer(k, np.uint8)
        d[i]["nonce"] = i
        d[i]["balance"] = np.frombuffer((100 + i).to_bytes(32, "big"), np.uint8)
        d[i]["code_hash"] = np.frombuffer(ke, np.uint8)
    s[0]["acct_key"] = np.frombuffer(k2, np.uint8)
    s[0]["slot_key"] = np.frombuffer(sk, np.uint8)
    s[0]["value"] = np.frombuffer((7).to_bytes(32, "big"), np.uint8)
    eng.apply_delta(d, s)
    accounts = {k1: (0, 100, ke, {}), k2: (1, 101, ke, {sk: 7})}
    assert eng.root() == bind.state_root(*to_arrays(accounts))


def test_overlay_rejects_slot_for_deleted_account(eng):
    acct, st = gen.gen_state_numpy(50, 2, bind.keccak256_batch)
    eng.upload(acct, st)
    k = bytes(acct[0]["key"])
    d = np.zeros(1, DELTA_DTYPE)
    d[0]["key"] = np.frombuffer(k, np.uint8)
    d[0]["deleted"] = 1
    s = np.zeros(1, bind.STORAGE_DTYPE)
    s[0]["acct_key"] = np.frombuffer(k, np.uint8)
    s[0]["slot_key"] = np.frombuffer(bind.keccak256(b"x"), np.uint8)
    s[0]["value"] = np.frombuffer((1).to_bytes(32, "big"), np.uint8)
    with pytest.raises(RuntimeError):
        eng.apply_delta(d, s)


# ---------------------------------------------------------------------------
# Dirty-path incremental (sre_root_retaining / sre_incremental_root):
# recompute only the 5-nibble cells a delta touches, reusing retained
# cell-top records. Parity vs the CPU oracle on dict-merged state across
# chained randomized delta sequences.
# ---------------------------------------------------------------------------

def _acct_dict(acct):
    return {bytes(a["key"]): [int(a["nonce"]),
                              int.from_bytes(bytes(a["balance"]), "big"),
                              bytes(a["code_hash"]), {}] for a in acct}


def _apply_dict(accounts, rows):
    for (k, n, b, ch, dead) in rows:
        if dead:
            accounts.pop(k, None)
        else:
            accounts[k] = [n, b, ch, {}]


def _want_root(accounts):
    return bind.state_root(*_arrays_of(accounts))


def test_incremental_chained_random_deltas(eng):
    rng = np.random.default_rng(1234)
    acct, _ = gen.gen_state_numpy(5000, 0, bind.keccak256_batch)
    accounts = _acct_dict(acct)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    assert eng.root_retaining() == _want_root(accounts)
    ke = bind.keccak256(b"")
    for step in range(6):
        keys = sorted(accounts)
        rows = []
        # modify a random subset
        for i in rng.choice(len(keys), size=40, replace=False):
            k = keys[int(i)]
            v = accounts[k]
            rows.append((k, v[0] + 1, v[1] + 7, v[2], 0))
        # delete a random subset (disjoint keys)
        dels = rng.choice(len(keys), size=15, replace=False)
        for i in dels:
            k = keys[int(i)]
            if any(r[0] == k for r in rows):
                continue
            rows.append((k, 0, 0, ke, 1))
        # insert fresh accounts
        for i in range(20):
            nk = bind.keccak256(b"inc" + bytes([step, i]))
            rows.append((nk, 3, 1000 + step * 100 + i, ke, 0))
        rows = sorted(set(rows))
        _apply_dict(accounts, rows)
        d, _ = _mk_delta(rows, [])
        assert eng.incremental_root(d) == _want_root(accounts), f"step {step}"


def test_incremental_adjacent_keys_same_cell(eng):
    # keys sharing long prefixes stress cell-boundary logic: several keys in
    # ONE 5-nibble cell, deltas that split/merge nodes inside it, plus a
    # neighbour cell that must be reused untouched.
    ke = bind.keccak256(b"")
    base = bytearray(bind.keccak256(b"cellbase"))
    keys = []
    for i in range(8):
        k = bytearray(base)
        k[31] = i  # same first 62 nibbles
        keys.append(bytes(k))
    other = bytearray(base)
    other[2] ^= 0x40  # different cell (3rd nibble differs)
    keys.append(bytes(other))
    accounts = {k: [1, 10 + i, ke, {}] for i, k in enumerate(sorted(keys))}
    eng.upload(*_arrays_of(accounts))
    assert eng.root_retaining() == _want_root(accounts)
    # delete half of the clustered keys, modify one, insert one deeper twin
    rows = []
    for k in sorted(keys)[:4]:
        rows.append((k, 0, 0, ke, 1))
    k5 = sorted(keys)[5]
    rows.append((k5, 9, 9999, ke, 0))
    twin = bytearray(base)
    twin[31] = 0xEE
    rows.append((bytes(twin), 2, 5, ke, 0))
    rows = sorted(rows)
    _apply_dict(accounts, rows)
    d, _ = _mk_delta(rows, [])
    assert eng.incremental_root(d) == _want_root(accounts)
    # second chained delta: re-insert one deleted key
    back = sorted(keys)[0]
    rows2 = [(back, 4, 44, ke, 0)]
    _apply_dict(accounts, rows2)
    d2, _ = _mk_delta(rows2, [])
    assert eng.incremental_root(d2) == _want_root(accounts)


def test_incremental_edge_cases(eng):
    ke = bind.keccak256(b"")
    acct, _ = gen.gen_state_numpy(300, 0, bind.keccak256_batch)
    accounts = _acct_dict(acct)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    r0 = eng.root_retaining()
    assert r0 == _want_root(accounts)
    # empty delta: same root
    assert eng.incremental_root(np.zeros(0, DELTA_DTYPE)) == r0
    # idempotent delta (rewrite same values)
    keys = sorted(accounts)
    rows = [(k, accounts[k][0], accounts[k][1], accounts[k][2], 0)
            for k in keys[:10]]
    d, _ = _mk_delta(rows, [])
    assert eng.incremental_root(d) == r0
    # delete everything -> EMPTY_ROOT, then refill from empty
    rows = [(k, 0, 0, ke, 1) for k in keys]
    d, _ = _mk_delta(rows, [])
    empty_root = bytes.fromhex(
        "56e81f171bcc55a6ff8345e692c0f86e5b48e01b996cadc001622fb5e363b421")
    assert eng.incremental_root(d) == empty_root
    # retention invalidated at na==0; re-arm and continue
    assert eng.root_retaining() == empty_root
    rows = [(bind.keccak256(b"solo"), 1, 2, ke, 0)]
    d, _ = _mk_delta(rows, [])
    accounts = {rows[0][0]: [1, 2, ke, {}]}
    assert eng.incremental_root(d) == _want_root(accounts)
    # single-account state (root is a leaf — coverage-hole fallback path)
    rows2 = [(bind.keccak256(b"solo2"), 7, 8, ke, 0)]
    accounts[rows2[0][0]] = [7, 8, ke, {}]
    d2, _ = _mk_delta(rows2, [])
    assert eng.incremental_root(d2) == _want_root(accounts)


def test_incremental_requires_retention(eng):
    acct, _ = gen.gen_state_numpy(100, 0, bind.keccak256_batch)
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    # no root_retaining yet -> error (upload invalidates retention)
    with pytest.raises(RuntimeError):
        eng.incremental_root(np.zeros(0, DELTA_DTYPE))


def test_incremental_with_storage_chained(eng):
    rng = np.random.default_rng(99)
    acct, st = gen.gen_state_numpy(2000, 4, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    assert eng.root_retaining() == bind.state_root(*_arrays_of(accounts))
    ke = bind.keccak256(b"")
    for step in range(4):
        keys = sorted(accounts)
        rows, strows = [], []
        # slot upserts + deletions on random surviving accounts
        for i in rng.choice(len(keys), 12, replace=False):
            k = keys[int(i)]
            slots = accounts[k][3]
            if slots and rng.random() < 0.4:
                dead = sorted(slots)[0]
                strows.append((k, dead, 0))
                del slots[dead]
            nk = bind.keccak256(b"slot" + bytes([step]) + k[:4])
            strows.append((k, nk, 1000 + step))
            slots[nk] = 1000 + step
        # delete one account WITH storage (wipe), insert one with storage
        victim = keys[int(rng.integers(len(keys)))]
        if not any(r[0] == victim for r in strows):
            rows.append((victim, 0, 0, ke, 1))
            del accounts[victim]
        nk = bind.keccak256(b"newacct" + bytes([step]))
        ns1 = bind.keccak256(b"ns" + bytes([step]))
        rows.append((nk, 1, 5, ke, 0))
        strows.append((nk, ns1, 77))
        accounts[nk] = [1, 5, ke, {ns1: 77}]
        # plain balance modification (storage must be KEPT)
        k2 = sorted(accounts)[3]
        if all(r[0] != k2 for r in rows):
            accounts[k2][1] += 9
            rows.append((k2, accounts[k2][0], accounts[k2][1],
                         accounts[k2][2], 0))
        rows = sorted(set(rows))
        strows = sorted(set(strows))
        d, s = _mk_delta(rows, strows)
        want = bind.state_root(*_arrays_of(accounts))
        assert eng.incremental_root(d, s) == want, f"step {step}"


def test_incremental_storage_wipe_all_slots(eng):
    ke = bind.keccak256(b"")
    ak = bind.keccak256(b"wipeme")
    slots = {bind.keccak256(b"s" + bytes([i])): i + 1 for i in range(5)}
    accounts = {ak: [3, 9, ke, dict(slots)],
                bind.keccak256(b"other"): [1, 1, ke, {}]}
    eng.upload(*_arrays_of(accounts))
    assert eng.root_retaining() == bind.state_root(*_arrays_of(accounts))
    # delete every slot -> storage root must become EMPTY_ROOT
    strows = [(ak, sk, 0) for sk in slots]
    accounts[ak][3].clear()
    d, s = _mk_delta([], strows)
    assert eng.incremental_root(d, s) == bind.state_root(*_arrays_of(accounts))
    # then re-add one
    sk = bind.keccak256(b"back")
    accounts[ak][3][sk] = 42
    d, s = _mk_delta([], [(ak, sk, 42)])
    assert eng.incremental_root(d, s) == bind.state_root(*_arrays_of(accounts))


def test_incremental_matches_apply_delta_path(eng):
    # same delta through (apply_delta + root) and incremental_root must agree
    rng = np.random.default_rng(77)
    acct, _ = gen.gen_state_numpy(2000, 0, bind.keccak256_batch)
    ke = bind.keccak256(b"")
    keys = sorted(bytes(a["key"]) for a in acct)
    rows = []
    for i in rng.choice(len(keys), size=25, replace=False):
        rows.append((keys[int(i)], 42, 4242, ke, 0))
    for i in range(5):
        rows.append((bind.keccak256(b"cmp" + bytes([i])), 1, i, ke, 0))
    rows = sorted(set(rows))
    d, _ = _mk_delta(rows, [])

    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    eng.apply_delta(d, np.zeros(0, bind.STORAGE_DTYPE))
    want = eng.root()

    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    eng.root_retaining()
    assert eng.incremental_root(d) == want


def test_incremental_storage_appears_after_drift(eng):
    # accounts-only base, several accounts-only deltas (fast path), THEN a
    # delta that introduces storage: retained roots indexing has drifted
    # but must not matter (all EMPTY while ns == 0)
    ke = bind.keccak256(b"")
    acct, _ = gen.gen_state_numpy(500, 0, bind.keccak256_batch)
    accounts = {bytes(a["key"]): [int(a["nonce"]),
                int.from_bytes(bytes(a["balance"]), "big"),
                bytes(a["code_hash"]), {}] for a in acct}
    eng.upload(acct, np.zeros(0, bind.STORAGE_DTYPE))
    assert eng.root_retaining() == bind.state_root(
        *_arrays_of({k: tuple(v) for k, v in accounts.items()}))
    for step in range(3):  # drift: inserts + deletes shift positions
        keys = sorted(accounts)
        rows = [(keys[step * 7], 0, 0, ke, 1)]
        del accounts[keys[step * 7]]
        nk = bind.keccak256(b"drift" + bytes([step]))
        rows.append((nk, 1, step, ke, 0))
        accounts[nk] = [1, step, ke, {}]
        d, _ = _mk_delta(sorted(rows), [])
        want = bind.state_root(
            *_arrays_of({k: tuple(v) for k, v in accounts.items()}))
        assert eng.incremental_root(d) == want
    # now storage appears
    tgt = sorted(accounts)[10]
    sk = bind.keccak256(b"first-slot")
    accounts[tgt][3][sk] = 123
    d, s = _mk_delta([], [(tgt, sk, 123)])
    want = bind.state_root(
        *_arrays_of({k: tuple(v) for k, v in accounts.items()}))
    assert eng.incremental_root(d, s) == want
    # and chains further with more storage
    sk2 = bind.keccak256(b"second-slot")
    accounts[tgt][3][sk2] = 9
    d, s = _mk_delta([], [(tgt, sk2, 9)])
    want = bind.state_root(
        *_arrays_of({k: tuple(v) for k, v in accounts.items()}))
    assert eng.incremental_root(d, s) == want
