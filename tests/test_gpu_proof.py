"""Account multiproof parity (sre_account_proof vs the oracle walker,
proof/mod.rs:59-137 semantics): per present target, identical root-first
node-RLP lists, plus an independent verify-by-replay property (walking the
proof from the root by target nibbles reaches the account leaf)."""
import numpy as np
import pytest

from oracle import bind, pyref
from reth_amd import gen
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
        accounts[bytes(a["key"])] = (int(a["nonce"]),
                                     int.from_bytes(bytes(a["balance"]), "big"),
                                     bytes(a["code_hash"]), {})
    for s in st:
        accounts[bytes(s["acct_key"])][3][bytes(s["slot_key"])] = \
            int.from_bytes(bytes(s["value"]), "big")
    return accounts


def _replay(nodes, key, root):
    """Walk the proof from the root following key nibbles; return the leaf
    value item payload. Minimal RLP walker, independent of the oracle."""
    def rlp_items(b):
        # parse one RLP list, return list of (payload bytes) items
        assert b[0] >= 0xC0
        if b[0] < 0xF8:
            pl, off = b[0] - 0xC0, 1
        else:
            n = b[0] - 0xF7
            pl, off = int.from_bytes(b[1:1 + n], "big"), 1 + n
        items, i = [], off
        end = off + pl
        while i < end:
            c = b[i]
            if c < 0x80:
                items.append(b[i:i + 1]); i += 1
            elif c < 0xB8:
                items.append(b[i + 1:i + 1 + c - 0x80]); i += 1 + c - 0x80
            elif c < 0xC0:
                n = c - 0xB7
                ln = int.from_bytes(b[i + 1:i + 1 + n], "big")
                items.append(b[i + 1 + n:i + 1 + n + ln]); i += 1 + n + ln
            else:  # nested list: keep raw encoding
                if c < 0xF8:
                    ln, hn = c - 0xC0, 1
                else:
                    hn = 1 + (c - 0xF7)
                    ln = int.from_bytes(b[i + 1:i + hn], "big")
                items.append(b[i:i + hn + ln]); i += hn + ln
        return items

    nib = []
    for byte in key:
        nib += [byte >> 4, byte & 0xF]
    by_hash = {bind.keccak256(n): n for n in nodes}
    assert root in by_hash, "first node must be the root"
    node = by_hash[root]
    pos = 0
    while True:
        items = rlp_items(node)
        if len(items) == 17:
            ref = items[nib[pos]]
            if not ref:
                return None  # empty child slot: absence proven
            pos += 1
            node = by_hash[ref] if len(ref) == 32 else ref
        else:
            hp = items[0]
            flag = hp[0] >> 4
            odd = flag & 1
            path = ([hp[0] & 0xF] if odd else [])
            for byte in hp[1:]:
                path += [byte >> 4, byte & 0xF]
            if nib[pos:pos + len(path)] != path:
                return None  # path divergence: absence proven
            pos += len(path)
            if flag & 2:  # leaf
                assert pos == 64
                return items[1]
            ref = items[1]
            node = by_hash[ref] if len(ref) == 32 else ref


def _check_state(eng, accounts, targets):
    acct, st = to_arrays(accounts)
    eng.upload(acct, st)
    root = bind.state_root(acct, st)
    assert eng.root() == root
    proofs = eng.account_proof(targets)
    for key, nodes in zip(targets, proofs):
        want = pyref.account_proof(accounts, key)
        assert nodes == want, f"proof mismatch for {key.hex()}"
        val = _replay(nodes, key, root)
        # leaf value = RLP([nonce, balance, storage_root, code_hash])
        nonce, bal, ch, slots = accounts[key]
        sr = pyref.storage_root(slots)
        assert val == pyref.account_value(nonce, bal, sr, ch)


def test_proof_random_state(eng):
    acct, st = gen.gen_state_numpy(3000, 4, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    keys = sorted(accounts)
    targets = [keys[0], keys[17], keys[1500], keys[-1]]
    _check_state(eng, accounts, targets)


def test_proof_multiproof_batch(eng):
    acct, st = gen.gen_state_numpy(20000, 0, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    keys = sorted(accounts)
    rng = np.random.default_rng(42)
    targets = [keys[int(i)] for i in rng.choice(len(keys), 16, replace=False)]
    _check_state(eng, accounts, targets)


def test_proof_small_tries(eng):
    ke = bind.keccak256(b"")
    # single account: proof = [leaf] with the full 64-nibble path
    k1 = bind.keccak256(b"one")
    _check_state(eng, {k1: (1, 2, ke, {})}, [k1])
    # two accounts: root branch (or ext+branch) + leaves
    k2 = bind.keccak256(b"two")
    acc = {k1: (1, 2, ke, {}), k2: (3, 4, ke, {})}
    _check_state(eng, acc, sorted(acc))
    # clustered keys sharing a long prefix: root extension node path
    base = bytearray(bind.keccak256(b"clu"))
    ks = []
    for i in range(4):
        b = bytearray(base); b[31] = i
        ks.append(bytes(b))
    acc = {k: (i, 10 + i, ke, {}) for i, k in enumerate(sorted(ks))}
    _check_state(eng, acc, sorted(acc))


def test_proof_absent_keys_exclusion(eng):
    acct, st = gen.gen_state_numpy(2000, 0, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    root = bind.state_root(acct, st)
    assert eng.root() == root
    absent = [bind.keccak256(b"definitely-absent" + bytes([i]))
              for i in range(6)]
    # mixed present/absent multiproof
    keys = sorted(accounts)
    targets = [keys[3]] + absent + [keys[-1]]
    proofs = eng.account_proof(targets)
    for k, nodes in zip(targets, proofs):
        assert nodes == pyref.account_proof(accounts, k), k.hex()
        got = _replay(nodes, k, root)
        if k in accounts:
            assert got is not None
        else:
            assert got is None  # replay must END at a proven divergence


def test_storage_proof_absent_exclusion(eng):
    acct, st = gen.gen_state_numpy(200, 8, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    assert eng.root() == bind.state_root(acct, st)
    keys = sorted(accounts)
    ak = keys[5]
    absents = [bind.keccak256(b"no-such-slot" + bytes([i])) for i in range(3)]
    pres = sorted(accounts[ak][3])[0]
    aks = [ak] * 4
    sks = absents + [pres]
    roots, proofs = eng.storage_proof(aks, sks)
    for sk, sr, nodes in zip(sks, roots, proofs):
        want_root, want_nodes = pyref.storage_proof(accounts, ak, sk)
        assert sr == want_root
        assert nodes == want_nodes, sk.hex()
        got = _replay(nodes, sk, sr)
        assert (got is None) == (sk not in accounts[ak][3])


def test_storage_proof_slotless_account(eng):
    ke = bind.keccak256(b"")
    a1, a2 = sorted([bind.keccak256(b"p"), bind.keccak256(b"q")])
    sk = bind.keccak256(b"s1")
    accounts = {a1: (1, 2, ke, {}), a2: (3, 4, ke, {sk: 9})}
    acct, st = to_arrays(accounts)
    eng.upload(acct, st)
    # a1 has no storage: exclusion proof of any slot is the empty list
    roots, proofs = eng.storage_proof([a1], [sk])
    assert roots[0] == bytes.fromhex(
        "56e81f171bcc55a6ff8345e692c0f86e5b48e01b996cadc001622fb5e363b421")
    assert proofs[0] == []


def test_storage_proof_absent_account(eng):
    # an ABSENT account behaves like a storage-less one:
    # StorageMultiProof::empty() — EMPTY_ROOT_HASH + empty node list
    # (proof/mod.rs storage_multiproof empty-cursor short circuit)
    ke = bind.keccak256(b"")
    a1 = bind.keccak256(b"present")
    sk = bind.keccak256(b"s1")
    accounts = {a1: (1, 2, ke, {sk: 5})}
    acct, st = to_arrays(accounts)
    eng.upload(acct, st)
    ghost = bind.keccak256(b"ghost-account")
    roots, proofs = eng.storage_proof([ghost, a1], [sk, sk])
    want_root, want_nodes = pyref.storage_proof(accounts, ghost, sk)
    assert roots[0] == want_root
    assert roots[0] == bytes.fromhex(
        "56e81f171bcc55a6ff8345e692c0f86e5b48e01b996cadc001622fb5e363b421")
    assert proofs[0] == [] and want_nodes == []
    # the present target in the same batch is unaffected
    want_root1, want_nodes1 = pyref.storage_proof(accounts, a1, sk)
    assert roots[1] == want_root1 and proofs[1] == want_nodes1


def test_storage_proof_parity(eng):
    acct, st = gen.gen_state_numpy(500, 24, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    root = bind.state_root(acct, st)
    assert eng.root() == root
    rng = np.random.default_rng(7)
    # pick (account, slot) pairs across several accounts
    aks, sks = [], []
    keys = sorted(accounts)
    for i in rng.choice(len(keys), 6, replace=False):
        ak = keys[int(i)]
        slots = sorted(accounts[ak][3])
        sks.append(slots[int(rng.integers(len(slots)))])
        aks.append(ak)
    roots, proofs = eng.storage_proof(aks, sks)
    for ak, sk, sr, nodes in zip(aks, sks, roots, proofs):
        want_root, want_nodes = pyref.storage_proof(accounts, ak, sk)
        assert sr == want_root
        assert nodes == want_nodes, f"storage proof mismatch {ak.hex()}/{sk.hex()}"
        # replay: walk from the storage root to the slot leaf
        val = _replay(nodes, sk, sr)
        assert val == pyref.rlp_int(accounts[ak][3][sk])


def test_storage_proof_small_values_inline_nodes(eng):
    # tiny values force inline (<32 B) deep nodes: exactly the case where
    # proof lists must SKIP embedded nodes
    ke = bind.keccak256(b"")
    ak = bind.keccak256(b"acct")
    slots = {}
    for i in range(40):
        slots[bind.keccak256(b"s" + bytes([i]))] = i + 1  # 1-byte values
    accounts = {ak: (1, 1, ke, slots)}
    acct, st = to_arrays(accounts)
    eng.upload(acct, st)
    assert eng.root() == bind.state_root(acct, st)
    sks = sorted(slots)[:5]
    roots, proofs = eng.storage_proof([ak] * 5, sks)
    for sk, sr, nodes in zip(sks, roots, proofs):
        want_root, want_nodes = pyref.storage_proof(accounts, ak, sk)
        assert sr == want_root
        assert nodes == want_nodes
