"""CPU-side checks of the proof oracle walkers (oracle/pyref.account_proof /
storage_proof): every proof verifies by replay against the trie root, with
inline-node skipping per eth_getProof semantics. (Engine parity is in
tests/test_gpu_proof.py.)"""
from oracle import bind, pyref
from reth_amd import gen
from tests.test_gpu_proof import _dict_of, _replay


def test_account_proofs_replay():
    acct, st = gen.gen_state_numpy(800, 3, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    root = bind.state_root(acct, st)
    keys = sorted(accounts)
    for k in [keys[0], keys[100], keys[-1]]:
        nodes = pyref.account_proof(accounts, k)
        assert bind.keccak256(nodes[0]) == root
        val = _replay(nodes, k, root)
        nonce, bal, ch, slots = accounts[k]
        assert val == pyref.account_value(nonce, bal,
                                          pyref.storage_root(slots), ch)


def test_storage_proofs_replay_incl_inline():
    ke = bind.keccak256(b"")
    slots = {bind.keccak256(bytes([i])): i + 1 for i in range(60)}
    accounts = {bind.keccak256(b"x"): (0, 1, ke, slots)}
    for sk in list(sorted(slots))[::17]:
        root, nodes = pyref.storage_proof(accounts, bind.keccak256(b"x"), sk)
        assert bind.keccak256(nodes[0]) == root
        assert _replay(nodes, sk, root) == pyref.rlp_int(slots[sk])


def test_absent_key_exclusion_replay():
    acct, st = gen.gen_state_numpy(50, 0, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    root = bind.state_root(acct, st)
    for i in range(5):
        k = bind.keccak256(b"nope" + bytes([i]))
        nodes = pyref.account_proof(accounts, k)
        assert nodes and bind.keccak256(nodes[0]) == root
        assert _replay(nodes, k, root) is None  # ends at proven divergence
    assert pyref.account_proof({}, bind.keccak256(b"x")) == []
