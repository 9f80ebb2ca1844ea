"""CPU oracle pinning: golden consensus vectors, KATs, pyref cross-checks.

These tests pin the CPU oracle (the parity checker for the HIP engine)
against everything that is available offline:
  - genesis stateRoot vectors (consensus-pinned) from
    /root/reference/crates/chainspec/res/genesis/*.json, committed as
    fixtures under tests/golden/
  - the hard-coded roots of /root/reference/crates/trie/db/tests/trie.rs:384-519
  - KECCAK_EMPTY / EMPTY_ROOT_HASH constants (trie.rs:15,775; merkle.rs:297)
  - random cross-checks between the two independent restatements
    (oracle/pyref.py dict-recursion vs oracle/mpt_oracle.c stream-recursion)
"""
import json
import os

import numpy as np
import pytest

from oracle import bind, pyref
from tests.util import GOLDEN, load_genesis, random_accounts, to_arrays

KECCAK_EMPTY = bytes.fromhex(
    "c5d2460186f7233c927e7db2dcc703c0e500b653ca82273b7bfad8045d85a470")
EMPTY_ROOT_HASH = bytes.fromhex(
    "56e81f171bcc55a6ff8345e692c0f86e5b48e01b996cadc001622fb5e363b421")


def test_keccak_kats():
    assert bind.keccak256(b"") == KECCAK_EMPTY
    assert bind.keccak256(b"\x80") == EMPTY_ROOT_HASH
    assert pyref.KECCAK_EMPTY == KECCAK_EMPTY
    assert pyref.EMPTY_ROOT_HASH == EMPTY_ROOT_HASH
    # multi-block + batch API vs pyref
    for msg in [b"a", b"abc", bytes(135), bytes(136), bytes(137), bytes(272),
                bytes(range(256)) * 3]:
        assert bind.keccak256(msg) == pyref.keccak256(msg)
    msgs = np.frombuffer(os.urandom(32 * 64), np.uint8).reshape(64, 32)
    got = bind.keccak256_batch(msgs)
    for i in range(64):
        assert bytes(got[i]) == pyref.keccak256(msgs[i].tobytes())


@pytest.mark.parametrize("name", ["mainnet", "sepolia", "holesky"])
def test_genesis_state_roots(name):
    accounts, want = load_genesis(name)
    acct, st = to_arrays(accounts)
    got = bind.state_root(acct, st)
    assert "0x" + got.hex() == want
    assert pyref.state_root(accounts) == got
    # shard-mode composition equals the monolithic root
    refs, lens, roots, counts = bind.subtree_roots(acct, st)
    assert bind.finish_top(refs, lens, roots, counts) == got


def test_fixed_vectors():
    fv = json.load(open(os.path.join(GOLDEN, "fixed_vectors.json")))
    # storage_root_regression (trie.rs:384): prehashed 4-slot vector
    sr = fv["storage_root_regression"]
    ak = bind.keccak256(bytes.fromhex("16b07afd1c635f77172e842a000ead9a2a222459"))
    st = np.zeros(4, dtype=bind.STORAGE_DTYPE)
    for i, (slot, v) in enumerate(sorted(sr["prehashed_slots"].items())):
        st[i]["acct_key"] = np.frombuffer(ak, np.uint8)
        st[i]["slot_key"] = np.frombuffer(bytes.fromhex(slot), np.uint8)
        st[i]["value"] = np.frombuffer(int(v, 16).to_bytes(32, "big"), np.uint8)
    acct = np.zeros(1, dtype=bind.ACCOUNT_DTYPE)
    acct[0]["key"] = np.frombuffer(ak, np.uint8)
    acct[0]["code_hash"] = np.frombuffer(KECCAK_EMPTY, np.uint8)
    roots = bind.storage_roots(acct, st)
    assert bytes(roots[0]).hex() == sr["storage_root"]

    # account_and_storage_trie (trie.rs:420-519): pinned root at :489
    av = fv["account_and_storage_trie"]
    items = {}
    for hk, nonce, bal, ch, srh in av["accounts"]:
        items[bytes.fromhex(hk)] = pyref.account_value(
            nonce, int(bal, 16),
            bytes.fromhex(srh) if srh else EMPTY_ROOT_HASH,
            bytes.fromhex(ch) if ch else KECCAK_EMPTY)
    assert pyref.trie_root(items).hex() == av["state_root"]


def test_empty_and_edge_cases():
    empty_a = np.zeros(0, dtype=bind.ACCOUNT_DTYPE)
    empty_s = np.zeros(0, dtype=bind.STORAGE_DTYPE)
    assert bind.state_root(empty_a, empty_s) == EMPTY_ROOT_HASH
    # single account
    accounts = random_accounts(trial=0, n=1)
    acct, st = to_arrays(accounts)
    assert bind.state_root(acct, st) == pyref.state_root(accounts)
    # single top nibble: root is not a branch; shard composition must agree
    accounts = random_accounts(trial=1, n=9, single_nibble=True)
    acct, st = to_arrays(accounts)
    want = pyref.state_root(accounts)
    assert bind.state_root(acct, st) == want
    refs, lens, roots, counts = bind.subtree_roots(acct, st)
    assert bind.finish_top(refs, lens, roots, counts) == want


def test_input_contract_rejected():
    accounts = random_accounts(trial=2, n=4)
    acct, st = to_arrays(accounts)
    bad = acct.copy()
    bad[[0, 1]] = bad[[1, 0]]  # unsorted accounts
    with pytest.raises(ValueError):
        bind.state_root(bad, st)
    if len(st):
        badst = st.copy()
        badst[0]["value"] = 0  # zero value must be absent
        with pytest.raises(ValueError):
            bind.state_root(acct, badst)
        orphan = st.copy()
        orphan[0]["acct_key"] = 0  # acct_key not in accounts
        with pytest.raises(ValueError):
            bind.state_root(acct, orphan)


@pytest.mark.parametrize("trial", range(12))
def test_random_cross_check(trial):
    n = [1, 2, 3, 5, 17, 64, 100, 150, 7, 33, 256, 40][trial]
    accounts = random_accounts(trial=trial + 100, n=n,
                               single_nibble=(trial % 5 == 0))
    acct, st = to_arrays(accounts)
    want = pyref.state_root(accounts)
    assert bind.state_root(acct, st) == want
    refs, lens, roots, counts = bind.subtree_roots(acct, st)
    assert bind.finish_top(refs, lens, roots, counts) == want


def test_state_root_par_matches_serial():
    """okc_state_root_par (the per-core CPU-baseline leg) is the same
    algorithm: identical roots on mixed shapes incl. single-nibble and
    empty-storage edges."""
    from reth_amd import gen
    acct, st = gen.gen_state_numpy(5000, 4, bind.keccak256_batch)
    assert bind.state_root_par(acct, st) == bind.state_root(acct, st)
    acct2, st2 = gen.gen_state_numpy(300, 0, bind.keccak256_batch)
    assert bind.state_root_par(acct2, st2) == bind.state_root(acct2, st2)
    assert bind.state_root_par(acct2[:1], st2) == bind.state_root(acct2[:1], st2)
