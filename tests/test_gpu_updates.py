"""TrieUpdates parity: engine rows must be byte-exact vs the CPU oracle
(which is itself pinned to the reference's hand-computed expectations in
tests/test_updates_cpu.py)."""
import numpy as np
import pytest

from oracle import bind
from reth_amd import gen
from tests.util import load_genesis, random_accounts, to_arrays

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()


def _compare(eng, acct, st):
    want_root, want_rows = bind.state_root_with_updates(acct, st)
    eng.upload(acct, st)
    got_root, got_rows = eng.root_with_updates()
    assert got_root == want_root
    assert len(got_rows) == len(want_rows), (len(got_rows), len(want_rows))
    ga = got_rows.view(np.uint8).reshape(len(got_rows), -1)
    wa = want_rows.view(np.uint8).reshape(len(want_rows), -1)
    if not np.array_equal(ga, wa):
        for i in range(len(got_rows)):
            if not np.array_equal(ga[i], wa[i]):
                raise AssertionError(
                    f"row {i}: engine={got_rows[i]} oracle={want_rows[i]}")


def test_dtype_matches_oracle():
    from reth_amd.engine import UPDATE_DTYPE
    assert UPDATE_DTYPE == bind.UPDATE_DTYPE


@pytest.mark.parametrize("na,slots", [(1000, 0), (500, 7), (2000, 16), (5000, 3)])
def test_generated_updates(eng, na, slots):
    acct, st = gen.gen_state_numpy(na, slots, bind.keccak256_batch)
    _compare(eng, acct, st)


@pytest.mark.parametrize("name", ["sepolia", "holesky", "mainnet"])
def test_genesis_updates(eng, name):
    accounts, _ = load_genesis(name)
    acct, st = to_arrays(accounts)
    _compare(eng, acct, st)


@pytest.mark.parametrize("trial", range(4))
def test_random_edge_updates(eng, trial):
    n = [2, 9, 64, 150][trial]
    accounts = random_accounts(trial=trial + 900, n=n,
                               single_nibble=(trial % 2 == 0))
    acct, st = to_arrays(accounts)
    _compare(eng, acct, st)
