"""GPU parity suite: the HIP engine must be bit-exact against the CPU oracle.

Everything here goes through the C-ABI (include/sre.h) via reth_amd.engine —
the same path a reth FFI binding would call. Golden fixtures (consensus
genesis roots) are committed under tests/golden; nothing reads
/root/reference at run time.
"""
import numpy as np
import pytest
import torch

from oracle import bind, pyref
from reth_amd import gen
from tests.util import load_genesis, random_accounts, to_arrays

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from reth_amd.engine import StateRootEngine
    e = StateRootEngine(0)
    yield e
    e.close()


def test_keccak_kernel_parity(eng):
    rng = np.random.default_rng(11)
    for msg_len in [1, 7, 8, 16, 20, 32, 55, 56, 64, 100, 135]:
        n = 4096
        msgs = rng.integers(0, 255, (n, msg_len), dtype=np.uint8)
        t = torch.from_numpy(msgs).cuda()
        out = torch.empty((n, 32), dtype=torch.uint8, device="cuda")
        eng.keccak_batch_device(t, msg_len, out)
        want = bind.keccak256_batch(msgs)
        got = out.cpu().numpy()
        assert np.array_equal(got, want), f"len={msg_len}"


@pytest.mark.parametrize("name", ["mainnet", "sepolia", "holesky"])
def test_genesis_roots_on_gpu(eng, name):
    accounts, want = load_genesis(name)
    acct, st = to_arrays(accounts)
    eng.upload(acct, st)
    got = eng.root()
    assert "0x" + got.hex() == want


@pytest.mark.parametrize("na,slots", [
    (1, 0), (1, 1), (2, 0), (3, 7), (100, 0), (1000, 0), (1000, 4),
    (200, 16), (5000, 2), (977, 31),
])
def test_generated_states(eng, na, slots):
    acct, st = gen.gen_state_numpy(na, slots, bind.keccak256_batch)
    want = bind.state_root(acct, st)
    eng.upload(acct, st)
    assert eng.root() == want
    # per-account storage roots
    got_roots = eng.storage_roots(len(acct))
    want_roots = bind.storage_roots(acct, st)
    assert np.array_equal(got_roots, want_roots)


@pytest.mark.parametrize("trial", range(8))
def test_random_edge_states(eng, trial):
    n = [1, 2, 3, 9, 64, 150, 256, 33][trial]
    accounts = random_accounts(trial=trial + 500, n=n,
                               single_nibble=(trial % 3 == 0))
    acct, st = to_arrays(accounts)
    want = bind.state_root(acct, st)
    eng.upload(acct, st)
    assert eng.root() == want


def test_subtree_composition(eng):
    acct, st = gen.gen_state_numpy(3000, 3, bind.keccak256_batch)
    want = bind.state_root(acct, st)
    eng.upload(acct, st)
    refs, lens, roots, counts = eng.subtree_roots()
    o_refs, o_lens, o_roots, o_counts = bind.subtree_roots(acct, st)
    assert np.array_equal(refs, o_refs) and np.array_equal(lens, o_lens)
    assert np.array_equal(roots, o_roots) and np.array_equal(counts, o_counts)
    assert eng.finish_top(refs, lens, roots, counts) == want
    # sharded-by-nibble composition: two half-shards combined == monolithic
    from reth_amd import sharding
    parts = []
    for r in range(2):
        a, s = gen.gen_state_numpy(3000, 3, bind.keccak256_batch,
                                   nibble_filter=lambda nib, rr=r: nib % 2 == rr)
        eng.upload(a, s)
        parts.append(eng.subtree_roots())
    m = sharding.combine(parts)
    assert eng.finish_top(*m) == want


def test_input_contract_rejected(eng):
    accounts = random_accounts(trial=901, n=6)
    acct, st = to_arrays(accounts)
    bad = acct.copy()
    bad[[0, 1]] = bad[[1, 0]]
    eng.upload(bad, st)
    with pytest.raises(RuntimeError):
        eng.root()
    if len(st):
        badst = st.copy()
        badst[0]["value"] = 0
        eng.upload(acct, badst)
        with pytest.raises(RuntimeError):
            eng.root()


def test_empty_state(eng):
    eng.upload(np.zeros(0, bind.ACCOUNT_DTYPE), np.zeros(0, bind.STORAGE_DTYPE))
    assert eng.root() == pyref.EMPTY_ROOT_HASH


def test_torch_generator_and_borrowed_tensors(eng):
    """GPU-generated state (device keccak + device sort) must equal the CPU
    generator, and the zero-copy borrow path must equal the upload path."""
    na, slots = 2000, 5
    acct_np, st_np = gen.gen_state_numpy(na, slots, bind.keccak256_batch)
    acct_t, st_t = gen.gen_state_torch(na, slots, eng.keccak_batch_device,
                                       device="cuda")
    a2, s2 = gen.tensors_to_np_state(acct_t, st_t)
    assert np.array_equal(acct_np.view(np.uint8).reshape(na, 104),
                          a2.view(np.uint8).reshape(na, 104))
    assert np.array_equal(st_np.view(np.uint8).reshape(-1, 96),
                          s2.view(np.uint8).reshape(-1, 96))
    want = bind.state_root(acct_np, st_np)
    eng.set_device_tensors(acct_t, st_t)
    assert eng.root() == want


def test_medium_state_50k(eng):
    """Larger parity point: 50k accounts x 16 slots (800k storage leaves)."""
    acct, st = gen.gen_state_numpy(50_000, 16, bind.keccak256_batch)
    want = bind.state_root(acct, st)
    eng.upload(acct, st)
    assert eng.root() == want
    s = eng.stats()
    assert s["leaf_count"] == len(st) + len(acct)
    assert s["branch_count"] > 0
