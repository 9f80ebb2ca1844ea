"""The two generator backends (numpy and torch) must produce identical bytes.

The torch backend runs on CPU here with the oracle's keccak standing in for
the device keccak kernel; on the GPU the same torch code runs with the HIP
kernel (covered by tests/test_gpu_parity.py).
"""
import numpy as np
import torch

from oracle import bind
from reth_amd import gen


def _keccak_cb_numpy(msgs):
    return bind.keccak256_batch(msgs)


def _keccak_cb_torch(in_u8, msg_len, out_u8):
    d = bind.keccak256_batch(in_u8.cpu().numpy()[:, :msg_len])
    out_u8.copy_(torch.from_numpy(d))
    return out_u8


def test_gen_backends_identical():
    for na, slots in [(1, 0), (7, 3), (64, 5), (200, 2)]:
        acct_np, st_np = gen.gen_state_numpy(na, slots, _keccak_cb_numpy)
        acct_t, st_t = gen.gen_state_torch(na, slots, _keccak_cb_torch,
                                           device="cpu")
        a2, s2 = gen.tensors_to_np_state(acct_t, st_t)
        assert np.array_equal(acct_np.view(np.uint8), a2.view(np.uint8))
        assert np.array_equal(st_np.view(np.uint8), s2.view(np.uint8))


def test_gen_shard_filter_partitions():
    na, slots = 300, 1
    acct_full, st_full = gen.gen_state_numpy(na, slots, _keccak_cb_numpy)
    parts_a, parts_s = [], []
    for rank in range(4):
        a, s = gen.gen_state_numpy(
            na, slots, _keccak_cb_numpy,
            nibble_filter=lambda nib, r=rank: (nib % 4) == r)
        parts_a.append(a)
        parts_s.append(s)
    assert sum(len(a) for a in parts_a) == len(acct_full)
    assert sum(len(s) for s in parts_s) == len(st_full)
    # union of shards equals the full set
    all_keys = np.concatenate([a["key"] for a in parts_a])
    full_keys = acct_full["key"]
    order = np.lexsort(tuple(all_keys[:, k] for k in reversed(range(32))))
    assert np.array_equal(all_keys[order], full_keys[
        np.lexsort(tuple(full_keys[:, k] for k in reversed(range(32))))])


def test_gen_values_exercise_rlp_branches():
    acct, st = gen.gen_state_numpy(50, 16, _keccak_cb_numpy)
    vals = np.asarray(st["value"])
    lens = 32 - np.argmax(vals != 0, axis=1)
    # minimal-BE lengths cover small and large values
    assert lens.min() >= 1
    assert lens.max() >= 28
    assert (lens == 1).any()
    # single-byte < 0x80 values occur (RLP self-encoding branch)
    small = [v for v in st["value"] if v[:31].sum() == 0 and v[31] < 0x80]
    assert len(small) > 0
    # entries strictly sorted by (acct_key, slot_key)
    raw = st.view(np.uint8).reshape(len(st), 96)[:, :64]
    for i in range(1, len(st)):
        assert raw[i - 1].tobytes() < raw[i].tobytes()
