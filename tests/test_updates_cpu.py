"""TrieUpdates (stored BranchNodeCompact rows) — oracle pinned against the
reference's own hand-computed expectations:

  - /root/reference/crates/trie/db/tests/trie.rs:519-540 (the 6-account
    fixture committed in tests/golden/fixed_vectors.json): exactly two
    account rows [0xB] and [0xB,0] with the asserted state/tree/hash masks.
  - /root/reference/crates/trie/trie/src/node_iter.rs:383-465: the 5-account
    zero-key fixture: rows at path 0*61 and 0*61+[1] with asserted masks.

Semantics (documented in include/sre.h): stored iff hash_mask != 0; hash bit
per branch child (>=32 B RLP); tree bit per stored child; root_hash only on
path-[] rows.
"""
import json
import os

import numpy as np

from oracle import bind, pyref
from tests.util import GOLDEN, load_genesis, to_arrays

KECCAK_EMPTY = pyref.KECCAK_EMPTY


def nibbles_of_row(r):
    pb = bytes(r["path"])
    return [(pb[k // 2] >> (0 if k & 1 else 4)) & 0xF
            for k in range(r["path_len"])]


def test_account_and_storage_trie_updates():
    fv = json.load(open(os.path.join(GOLDEN, "fixed_vectors.json")))
    av = fv["account_and_storage_trie"]
    sr = fv["storage_root_regression"]
    accounts = {}
    for hk, nonce, bal, ch, srh in av["accounts"]:
        accounts[bytes.fromhex(hk)] = (
            nonce, int(bal, 16),
            bytes.fromhex(ch) if ch else KECCAK_EMPTY, {})
    acct, _ = to_arrays(accounts)
    ak3 = bind.keccak256(bytes.fromhex("16b07afd1c635f77172e842a000ead9a2a222459"))
    strows = [(ak3, bytes.fromhex(s), int(v, 16))
              for s, v in sorted(sr["prehashed_slots"].items())]
    st = np.zeros(len(strows), dtype=bind.STORAGE_DTYPE)
    for i, (a, s, v) in enumerate(strows):
        st[i]["acct_key"] = np.frombuffer(a, np.uint8)
        st[i]["slot_key"] = np.frombuffer(s, np.uint8)
        st[i]["value"] = np.frombuffer(v.to_bytes(32, "big"), np.uint8)

    root, rows = bind.state_root_with_updates(acct, st)
    assert root.hex() == av["state_root"]
    arows = rows[rows["kind"] == 0]
    # pinned: trie.rs:521-540
    assert len(arows) == 2
    r0, r1 = arows[0], arows[1]
    assert nibbles_of_row(r0) == [0xB]
    assert (r0["state_mask"], r0["tree_mask"], r0["hash_mask"]) == (0b1011, 0b0001, 0b1001)
    assert r0["num_hashes"] == 2 and r0["root_hash_set"] == 0
    assert nibbles_of_row(r1) == [0xB, 0x0]
    assert (r1["state_mask"], r1["tree_mask"], r1["hash_mask"]) == (0b10001, 0, 0b10000)
    assert r1["num_hashes"] == 1 and r1["root_hash_set"] == 0
    # the 4-slot storage trie: root branch (children 1, 3); child 1 is a
    # hashed branch, child 3 an INLINE branch (26 B) => no hash bit
    srows = rows[rows["kind"] == 1]
    assert len(srows) == 1
    s0 = srows[0]
    assert bytes(s0["acct_key"]) == ak3
    assert s0["path_len"] == 0 and s0["root_hash_set"] == 1
    assert (s0["state_mask"], s0["tree_mask"], s0["hash_mask"]) == (0b1010, 0, 0b10)
    assert bytes(s0["root_hash"]).hex() == sr["storage_root"]


def test_node_iter_fixture_updates():
    keys = sorted(bytes.fromhex(h) for h in [
        "0000000000000000000000000000000000000000000000000000000000000000",
        "0000000000000000000000000000000000000000000000000000000000000010",
        "0000000000000000000000000000000000000000000000000000000000000100",
        "0000000000000000000000000000000000000000000000000000000000000101",
        "0000000000000000000000000000000000000000000000000000000000000110"])
    acct = np.zeros(5, dtype=bind.ACCOUNT_DTYPE)
    for i, k in enumerate(keys):
        acct[i]["key"] = np.frombuffer(k, np.uint8)
        acct[i]["code_hash"] = np.frombuffer(KECCAK_EMPTY, np.uint8)
    root, rows = bind.state_root_with_updates(acct, np.zeros(0, bind.STORAGE_DTYPE))
    arows = rows[rows["kind"] == 0]
    # pinned: node_iter.rs:437-460 (branch_node_0 and branch_node_2)
    assert len(arows) == 2
    b0, b2 = arows[0], arows[1]
    assert nibbles_of_row(b0) == [0] * 61
    assert (b0["state_mask"], b0["tree_mask"], b0["hash_mask"]) == (0b11, 0b10, 0b11)
    assert b0["num_hashes"] == 2 and b0["root_hash_set"] == 0
    assert nibbles_of_row(b2) == [0] * 61 + [1]
    assert (b2["state_mask"], b2["tree_mask"], b2["hash_mask"]) == (0b11, 0b00, 0b01)
    assert b2["num_hashes"] == 1 and b2["root_hash_set"] == 0


def test_updates_invariants_on_genesis():
    accounts, want = load_genesis("holesky")
    acct, st = to_arrays(accounts)
    root, rows = bind.state_root_with_updates(acct, st)
    assert "0x" + root.hex() == want
    assert len(rows) > 0
    for r in rows:
        assert r["hash_mask"] != 0          # stored iff hash_mask != 0
        assert r["num_hashes"] == bin(int(r["hash_mask"])).count("1")
        # tree_mask and hash_mask are subsets of state_mask
        assert (int(r["tree_mask"]) & ~int(r["state_mask"])) == 0
        assert (int(r["hash_mask"]) & ~int(r["state_mask"])) == 0
        assert r["root_hash_set"] == (1 if r["path_len"] == 0 else 0)
    # rows sorted: account rows first, each list path-sorted
    kinds = [int(r["kind"]) for r in rows]
    assert kinds == sorted(kinds)
