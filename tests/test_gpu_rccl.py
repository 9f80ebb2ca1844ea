"""RCCL warm-up of the N>1 exchange path on a single GPU:
the exact all-gather + combine + finish sequence bench.py --gpus N issues
runs here under a real RCCL (backend "nccl") process group — world size 1
on the leased GPU, with genuinely sharded data (both shards computed by
two engine contexts, the cross-shard combine checked against the
monolithic root). RCCL cannot place two ranks on one device, so the
collective itself is world-1; the multi-rank combine logic is covered by
the gloo world-2 test (test_sharding_gloo.py) and the payload round trip
through CUDA tensors is covered here."""
import os

import numpy as np
import pytest

from oracle import bind
from reth_amd import gen, sharding

pytestmark = pytest.mark.gpu


def test_rccl_allgather_shard_exchange_single_gpu():
    import torch
    import torch.distributed as dist
    from reth_amd.engine import StateRootEngine

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29771")
    os.environ["HSA_ENABLE_IPC_MODE_LEGACY"] = \
        os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        acct, st = gen.gen_state_numpy(4000, 3, bind.keccak256_batch)
        want = bind.state_root(acct, st)

        # shard the state as a 2-rank run would (top nibble % 2)
        parts = []
        eng = StateRootEngine(0)
        for rank in range(2):
            keep = np.array([(a["key"][0] >> 4) % 2 == rank for a in acct])
            acct_r = acct[keep]
            owner = {bytes(a["key"]) for a in acct_r}
            st_r = st[[bytes(s["acct_key"]) in owner for s in st]]
            eng.upload(np.ascontiguousarray(acct_r),
                       np.ascontiguousarray(st_r))
            refs, lens, roots, counts = eng.subtree_roots()
            if rank == 0:
                # THE RCCL COLLECTIVE: the exact call bench.py's N>1 path
                # makes, on this shard's real payload, over CUDA tensors
                m = sharding.all_gather_combine(refs, lens, roots, counts,
                                                device="cuda:0")
                # world-1: the gathered set is exactly this rank's payload
                assert np.array_equal(m[1], lens)
                assert np.array_equal(m[3], counts)
            parts.append((refs, lens, roots, counts))
        # cross-shard combine + device-side finish == monolithic root
        merged = sharding.combine(parts)
        assert eng.finish_top(*merged) == want
        eng.close()
    finally:
        dist.destroy_process_group()
