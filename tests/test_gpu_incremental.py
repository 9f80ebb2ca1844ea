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
