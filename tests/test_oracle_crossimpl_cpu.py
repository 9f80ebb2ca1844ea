"""Cross-implementation oracle fuzz (CPU): the C restatement
(oracle/mpt_oracle.c) and the independent pure-python reference
(oracle/pyref.py) must agree on roots, storage roots and proofs for
randomized states — two restatements of the reference algorithm written
against different sources (in-repo proof_v2 shapes vs the yellow-paper
spec), so agreement pins both."""
import numpy as np

from oracle import bind, pyref
from tests.test_gpu_fuzz import _rand_state
from tests.util import to_arrays


def test_roots_agree_randomized():
    rng = np.random.default_rng(0x5EED5)
    for _ in range(8):
        na = int(rng.integers(1, 300))
        slots = int(rng.integers(0, 5))
        accounts = _rand_state(rng, na, slots)
        acct, st = to_arrays(accounts)
        assert bind.state_root(acct, st) == pyref.state_root(accounts)
        assert bind.state_root_par(acct, st) == pyref.state_root(accounts)


def test_storage_roots_agree():
    rng = np.random.default_rng(0xA11CE)
    accounts = _rand_state(rng, 40, 6)
    acct, st = to_arrays(accounts)
    roots = bind.storage_roots(acct, st)
    for i, a in enumerate(acct):
        k = bytes(a["key"])
        assert bytes(roots[i]) == pyref.storage_root(accounts[k][3])
