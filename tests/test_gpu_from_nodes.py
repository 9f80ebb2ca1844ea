"""sre_root_from_nodes parity — the state_root_from_nodes / TrieInput
surface (crates/storage/storage-api/src/trie.rs:26-40): roots computed
with a stored-node overlay must be bit-exact, and supplied kind-1
path-[] rows must actually short-circuit untouched storage tries
(checked through engine stats: the storage leaf work collapses to the
touched/row-less tries)."""
import numpy as np
import pytest

from oracle import bind
from reth_amd import gen
from reth_amd.engine import DELTA_DTYPE
from tests.test_gpu_incremental import (_dict_of, _arrays_of, _mk_delta)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()


def test_round_trip_rows_reproduce_root(eng):
    # rows from root_with_updates fed back with an empty delta must
    # reproduce the root WITHOUT redoing the storage tries that have rows
    acct, st = gen.gen_state_numpy(3000, 8, bind.keccak256_batch)
    eng.upload(acct, st)
    root, rows = eng.root_with_updates()
    full_leaves = eng.stats()["leaf_count"]

    eng.upload(acct, st)
    assert eng.root_from_nodes(rows) == root
    seeded_leaves = eng.stats()["leaf_count"]
    # with 8 uniform slots some tries' root branches are unstored (no
    # row) and get rebuilt; the bulk must be seeded away
    assert seeded_leaves < full_leaves / 2, (seeded_leaves, full_leaves)


def test_from_nodes_with_delta(eng):
    rng = np.random.default_rng(31)
    acct, st = gen.gen_state_numpy(2000, 6, bind.keccak256_batch)
    accounts = _dict_of(acct, st)
    eng.upload(acct, st)
    root0, rows = eng.root_with_updates()
    ke = bind.keccak256(b"")
    keys = sorted(accounts)
    rowsd, strows = [], []
    # touch storage of 5 accounts
    for i in rng.choice(len(keys), 5, replace=False):
        k = keys[int(i)]
        nk = bind.keccak256(b"fn" + k[:4])
        strows.append((k, nk, 1234))
        accounts[k][3][nk] = 1234
    # modify 10 accounts (storage untouched -> must come from rows)
    for i in rng.choice(len(keys), 10, replace=False):
        k = keys[int(i)]
        if any(r[0] == k for r in strows):
            continue
        accounts[k][1] += 5
        rowsd.append((k, accounts[k][0], accounts[k][1], accounts[k][2], 0))
    # delete 3 accounts (incl. their storage)
    ndel = 0
    for i in rng.choice(len(keys), 10, replace=False):
        k = keys[int(i)]
        if ndel >= 3 or any(r[0] == k for r in strows) or \
                any(r[0] == k for r in rowsd):
            continue
        rowsd.append((k, 0, 0, ke, 1))
        del accounts[k]
        ndel += 1
    # new account with storage
    nk = bind.keccak256(b"fn-new")
    ns = bind.keccak256(b"fn-slot")
    rowsd.append((nk, 1, 2, ke, 0))
    strows.append((nk, ns, 7))
    accounts[nk] = [1, 2, ke, {ns: 7}]

    d, s = _mk_delta(sorted(set(rowsd)), sorted(set(strows)))
    eng.upload(acct, st)
    got = eng.root_from_nodes(rows, d, s)
    assert got == bind.state_root(*_arrays_of(accounts))


def test_from_nodes_storage_becomes_empty(eng):
    # delta deletes every slot of one account: its supplied root row must
    # NOT be used (touched), and the account leaf reverts to EMPTY_ROOT
    ke = bind.keccak256(b"")
    ak = bind.keccak256(b"fe-acct")
    slots = {bind.keccak256(b"fe" + bytes([i])): i + 1 for i in range(6)}
    accounts = {ak: [1, 1, ke, dict(slots)],
                bind.keccak256(b"fe-other"): [2, 2, ke, {}]}
    eng.upload(*_arrays_of(accounts))
    root0, rows = eng.root_with_updates()
    strows = [(ak, sk, 0) for sk in slots]
    accounts[ak][3].clear()
    d, s = _mk_delta([], sorted(strows))
    eng.upload(*_arrays_of({ak: (1, 1, ke, dict(slots)),
                            bind.keccak256(b"fe-other"): (2, 2, ke, {})}))
    got = eng.root_from_nodes(rows, d, s)
    assert got == bind.state_root(*_arrays_of(accounts))


def test_from_nodes_rejects_removal_rows(eng):
    from reth_amd.engine import UPDATE_DTYPE
    acct, st = gen.gen_state_numpy(100, 2, bind.keccak256_batch)
    eng.upload(acct, st)
    bad = np.zeros(1, dtype=UPDATE_DTYPE)
    bad[0]["removed"] = 1
    with pytest.raises(RuntimeError):
        eng.root_from_nodes(bad)


def test_from_nodes_empty_rows_is_plain_root(eng):
    acct, st = gen.gen_state_numpy(500, 3, bind.keccak256_batch)
    eng.upload(acct, st)
    want = eng.root()
    eng.upload(acct, st)
    from reth_amd.engine import UPDATE_DTYPE
    assert eng.root_from_nodes(np.zeros(0, dtype=UPDATE_DTYPE)) == want
