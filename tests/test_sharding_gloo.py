"""Multi-process (gloo, world_size=2) test of the sharding decomposition.

Each rank generates its nibble shard, computes per-nibble subtrie digests
with the CPU oracle, exchanges them through reth_amd.sharding's all-gather
path (the same code bench.py uses over RCCL), and finishes the root. The
result must equal the monolithic oracle root of the full state.
"""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from oracle import bind
from reth_amd import gen, sharding

N_ACCOUNTS = 400
SLOTS = 3


def _rank_main(rank, world, port, q):
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank,
        world_size=world)
    try:
        acct, st = gen.gen_state_numpy(
            N_ACCOUNTS, SLOTS, bind.keccak256_batch,
            nibble_filter=lambda nib: (nib % world) == rank)
        refs, lens, roots, counts = bind.subtree_roots(acct, st)
        m_refs, m_lens, m_roots, m_counts = sharding.all_gather_combine(
            refs, lens, roots, counts, device="cpu")
        root = bind.finish_top(m_refs, m_lens, m_roots, m_counts)
        q.put((rank, root))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_sharded_root_matches_monolithic():
    world = 2
    port = 29511
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, root = q.get(timeout=240)
        results[rank] = root
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # both ranks agree
    assert results[0] == results[1]
    # and match the monolithic oracle root
    acct, st = gen.gen_state_numpy(N_ACCOUNTS, SLOTS, bind.keccak256_batch)
    assert results[0] == bind.state_root(acct, st)


def test_combine_rejects_double_ownership():
    refs = np.zeros((16, 33), np.uint8)
    lens = np.zeros(16, np.uint8)
    roots = np.zeros((16, 32), np.uint8)
    counts = np.zeros(16, np.uint64)
    lens[3] = 33
    with pytest.raises(RuntimeError):
        sharding.combine([(refs, lens, roots, counts)] * 2)


def test_pack_unpack_roundtrip():
    rng = np.random.default_rng(3)
    refs = rng.integers(0, 255, (16, 33), dtype=np.uint8)
    lens = rng.integers(0, 34, 16, dtype=np.uint8)
    roots = rng.integers(0, 255, (16, 32), dtype=np.uint8)
    counts = rng.integers(0, 1 << 40, 16, dtype=np.uint64)
    r, l, ro, c = sharding.unpack(sharding.pack(refs, lens, roots, counts))
    assert np.array_equal(r, refs) and np.array_equal(l, lens)
    assert np.array_equal(ro, roots) and np.array_equal(c, counts)
