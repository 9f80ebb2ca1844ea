"""Multi-GPU sharding of the account trie by hashed-key top nibble.

Decomposition: storage tries are independent per account and
the account trie is a prefix trie, so rank r owns every account whose hashed
key's top nibble n satisfies n % world == r. Each rank computes its storage
roots + per-nibble subtrie digests (sre_subtree_roots); one all-gather of the
16 x (child_ref, root_hash, count) records (~1.2 KB per rank, latency-bound)
over RCCL/xGMI; every rank finishes the root branch locally (sre_finish_top).

This module is backend-agnostic (numpy payloads + torch.distributed): the
gloo CPU test drives it with the oracle, bench.py with the HIP engine.
"""
import numpy as np

PAYLOAD_BYTES = 16 * 33 + 16 + 16 * 32 + 16 * 8  # refs, lens, roots, counts


def nibble_filter(rank, world):
    return lambda nib: (nib % world) == rank


def pack(refs, lens, roots, counts) -> np.ndarray:
    buf = np.zeros(PAYLOAD_BYTES, dtype=np.uint8)
    o = 0
    for part in (np.ascontiguousarray(refs, np.uint8).ravel(),
                 np.ascontiguousarray(lens, np.uint8).ravel(),
                 np.ascontiguousarray(roots, np.uint8).ravel(),
                 np.ascontiguousarray(counts, np.uint64).view(np.uint8).ravel()):
        buf[o:o + len(part)] = part
        o += len(part)
    assert o == PAYLOAD_BYTES
    return buf


def unpack(buf: np.ndarray):
    o = 0
    refs = buf[o:o + 16 * 33].reshape(16, 33).copy(); o += 16 * 33
    lens = buf[o:o + 16].copy(); o += 16
    roots = buf[o:o + 16 * 32].reshape(16, 32).copy(); o += 16 * 32
    counts = buf[o:o + 16 * 8].copy().view(np.uint64); o += 16 * 8
    return refs, lens, roots, counts


def combine(parts):
    """parts: list of (refs, lens, roots, counts), one per rank. Each nibble
    must be owned by at most one rank. Returns merged (refs,lens,roots,counts)."""
    refs = np.zeros((16, 33), dtype=np.uint8)
    lens = np.zeros(16, dtype=np.uint8)
    roots = np.zeros((16, 32), dtype=np.uint8)
    counts = np.zeros(16, dtype=np.uint64)
    for r, l, ro, cn in parts:
        for b in range(16):
            if l[b]:
                if lens[b]:
                    raise RuntimeError(f"nibble {b} produced by two ranks")
                refs[b] = r[b]
                lens[b] = l[b]
                roots[b] = ro[b]
            counts[b] += cn[b]
    return refs, lens, roots, counts


def all_gather_combine(refs, lens, roots, counts, device="cpu"):
    """torch.distributed all-gather of this rank's subtree payload; returns
    the merged (refs, lens, roots, counts). Works over gloo (cpu) and
    nccl/RCCL (cuda tensors)."""
    import torch
    import torch.distributed as dist
    world = dist.get_world_size()
    local = torch.from_numpy(pack(refs, lens, roots, counts)).to(device)
    bufs = [torch.empty_like(local) for _ in range(world)]
    dist.all_gather(bufs, local)
    parts = [unpack(b.cpu().numpy()) for b in bufs]
    return combine(parts)
