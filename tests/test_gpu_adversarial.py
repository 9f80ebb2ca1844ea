"""Adversarial trie topologies: deep extensions, ragged storage, dense
sibling fans — shapes the uniform random generator rarely produces.

All states are built CPU-side with crafted keys, checked bit-exact:
engine (C-ABI) == C oracle == (transitively) pyref.
"""
import numpy as np
import pytest

from oracle import bind
from tests.util import to_arrays

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()


def _acct(key: bytes, nonce=1, bal=10**18, slots=None):
    return key, (nonce, bal,
                 bind.keccak256(b""),  # KECCAK_EMPTY
                 slots or {})


def _check(eng, accounts):
    acct, st = to_arrays(dict(accounts))
    want = bind.state_root(acct, st)
    eng.upload(acct, st)
    assert eng.root() == want
    refs, lens, roots, counts = eng.subtree_roots()
    o = bind.subtree_roots(acct, st)
    assert np.array_equal(refs, o[0]) and np.array_equal(lens, o[1])
    assert eng.finish_top(refs, lens, roots, counts) == want


def test_deep_shared_prefixes(eng):
    """Keys agreeing on 40-63 nibbles: branches at extreme depths, long
    extension nodes, 1-nibble short keys."""
    base = bind.keccak256(b"deep")
    accounts = []
    for last in [0x00, 0x01, 0x0F, 0x10, 0xFF]:
        accounts.append(_acct(base[:31] + bytes([last])))       # diverge at nibble 62/63
    for mid in [0x00, 0x80]:
        accounts.append(_acct(base[:20] + bytes([mid]) + base[21:]))  # nibble 40
    _check(eng, accounts)


def test_full_sibling_fan(eng):
    """A branch with all 16 children at several depths."""
    base = bind.keccak256(b"fan")
    accounts = []
    for n0 in range(16):  # all 16 children of the root branch
        accounts.append(_acct(bytes([(n0 << 4) | (base[0] & 0xF)]) + base[1:]))
    for n1 in range(16):  # all 16 children of a depth-30 branch
        k = bytearray(base)
        k[15] = (k[15] & 0xF0) | n1
        accounts.append(_acct(bytes(k)))
    _check(eng, accounts)


def test_ragged_storage(eng):
    """Accounts with 0 / 1 / 2 / 333 slots, plus deep-prefix slot keys."""
    accounts = []
    for i in range(12):
        key = bind.keccak256(b"rag" + bytes([i]))
        slots = {}
        if i % 4 == 1:
            slots = {bind.keccak256(b"s" + bytes([i])): 1}  # 1-slot trie
        elif i % 4 == 2:
            sbase = bind.keccak256(b"t" + bytes([i]))
            # two slots diverging at the last nibble: branch at depth 63
            slots = {sbase[:31] + b"\x00": 0x7F, sbase[:31] + b"\x01": 0x80}
        elif i % 4 == 3:
            slots = {bind.keccak256(bytes([i]) + j.to_bytes(2, "little")): j + 1
                     for j in range(333)}
        accounts.append(_acct(key, slots=slots))
    _check(eng, accounts)


def test_value_rlp_boundaries(eng):
    """Storage values at every RLP encoding boundary."""
    key = bind.keccak256(b"vals")
    vals = [1, 0x7F, 0x80, 0xFF, 0x100, 0xFFFF, 0x10000,
            (1 << 55) - 1, 1 << 55, (1 << 64) - 1, 1 << 64,
            (1 << 248) | 5, (1 << 256) - 1]
    slots = {bind.keccak256(b"v" + bytes([i])): v for i, v in enumerate(vals)}
    accounts = [_acct(key, slots=slots),
                _acct(bind.keccak256(b"other"), nonce=0, bal=0x7F),
                _acct(bind.keccak256(b"rich"), nonce=(1 << 64) - 1,
                      bal=(1 << 256) - 1)]
    _check(eng, accounts)


def test_inline_node_chains(eng):
    """Tiny values + deep divergence: inline (<32 B) leaf and branch nodes
    embedded in parents rather than hashed."""
    accounts = []
    key = bind.keccak256(b"inline")
    sbase = bind.keccak256(b"islots")
    slots = {}
    # slots diverging at depth 60+: short keys of 1-3 nibbles, 1-byte values
    for i in range(6):
        k = bytearray(sbase)
        k[30] = i
        k[31] = (7 * i + 3) % 256
        slots[bytes(k)] = i + 1
    accounts.append(_acct(key, slots=slots))
    _check(eng, accounts)


def test_medium_64slot_parity(eng):
    """Bit-exact parity at a 64-slot shape (the bench configuration's
    per-account structure) on 5k accounts: 320k storage leaves."""
    from reth_amd import gen
    acct, st = gen.gen_state_numpy(5000, 64, bind.keccak256_batch)
    want = bind.state_root(acct, st)
    eng.upload(acct, st)
    assert eng.root() == want
