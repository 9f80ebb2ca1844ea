#!/usr/bin/env python3
"""Benchmark: state-root wall-clock on the BASELINE.json workload.

A "step" = one full state-root build (storage tries + account trie) over a
synthetic HashedPostState of the named shape, inputs already resident in HBM
when the timed region starts (generation + upload are untimed setup).

Single GPU (default): the largest single-GPU configuration of the metric —
10M accounts x 64 slots — via sre_root.
Multi GPU (launched by the driver via torch.distributed.run, one rank per
GPU over RCCL): accounts sharded by hashed-key top nibble (nibble % world ==
rank); per-rank sre_subtree_roots, one all-gather of the 16 subtrie digests,
root finished on every rank (reth_amd/sharding.py). scaling="strong": the
TOTAL job (one root over the same 10M x 64 state) is fixed as ranks grow;
the driver computes efficiency from the per-N values.

CPU baseline: the C oracle (reth-algorithm restatement), one thread per host
core, timed on the ACTUAL full-shape state (the same device-generated
entries, copied to host; oracle root asserted equal to the GPU root) —
measured, not extrapolated. The single-thread figure is sample-scaled and
labeled as such. Both are reported baselines, not the optimisation target.
"""
import argparse
import json
import os
import sys
import time


def _pmc_traffic(accounts, slots):
    """PMC-measured HBM bytes (GB) per leaf-kernel launch for this shape,
    from the committed rocprofv3 counter runs (profiles/traffic.json);
    None when the shape was not profiled."""
    try:
        with open(os.path.join(os.path.dirname(os.path.abspath(__file__)),
                               "profiles", "traffic.json")) as f:
            t = json.load(f)
        return t.get(f"{accounts}x{slots}", {}).get("leaf_traffic_gb")
    except Exception:
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--accounts", type=int, default=10_000_000)
    ap.add_argument("--slots", type=int, default=64)
    ap.add_argument("--cpu-sample-accounts", type=int, default=60_000)
    ap.add_argument("--cpu-baseline", choices=["full", "sample", "off"],
                    default="full",
                    help="full: run the threaded C oracle on the ACTUAL "
                         "full-shape state (D2H of the same device-generated "
                         "entries; ~3 min at 10M x 64, root checked against "
                         "the GPU result); sample: bounded sample scaled "
                         "linearly in leaves; off: skip")
    ap.add_argument("--no-cpu-baseline", action="store_true",
                    help="alias for --cpu-baseline off")
    ap.add_argument("--check", action="store_true",
                    help="extra property check: sharded composition == root")
    ap.add_argument("--incremental", action="store_true",
                    help="BASELINE configs[4]: N-account resident base + "
                         "delta-accounts overlay delta per step (N=1 only)")
    ap.add_argument("--delta-accounts", type=int, default=5000)
    ap.add_argument("--dirty", action="store_true",
                    help="with --incremental: dirty-path recompute "
                         "(sre_incremental_root) instead of merge+full root")
    ap.add_argument("--incremental-slots", type=int, default=0,
                    help="with --incremental: slots/account in the resident "
                         "base, plus per-step storage delta rows "
                         "(0 = the configs[4] accounts-only shape)")
    args = ap.parse_args()
    if args.incremental:
        args.slots = args.incremental_slots  # configs[4] default: accounts-only

    import torch

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if args.gpus > 1 and world == 1:
        print("--gpus > 1 requires torch.distributed.run (one rank per GPU)",
              file=sys.stderr)
        sys.exit(2)
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        dist.init_process_group("nccl")
        torch.cuda.set_device(local_rank)

    if not torch.cuda.is_available():
        print("bench.py requires an MI355X (no CPU fallback)", file=sys.stderr)
        sys.exit(1)

    from reth_amd import gen, sharding
    from reth_amd.engine import StateRootEngine

    eng = StateRootEngine(local_rank)

    # ---- untimed setup: generate on-GPU, borrow tensors (zero-copy) ----
    nf = sharding.nibble_filter(rank, world) if world > 1 else None
    t_gen0 = time.perf_counter()
    acct_t, st_t = gen.gen_state_torch(args.accounts, args.slots,
                                       eng.keccak_batch_device,
                                       device=f"cuda:{local_rank}",
                                       nibble_filter=nf)
    torch.cuda.synchronize()
    t_gen = time.perf_counter() - t_gen0
    na, ns = acct_t.shape[0], st_t.shape[0]
    eng.set_device_tensors(acct_t, st_t)
    # release torch's cached generation blocks back to HIP so the engine's
    # own allocations (records, level buffers) don't OOM at the 10M x 64 size
    torch.cuda.empty_cache()

    # ---- incremental mode (configs[4]): build the idempotent overlay delta
    # (absolute-valued modifications / inserts / deletes => every step
    # produces the same post-delta root, so the determinism check holds)
    delta = None
    if args.incremental:
        if world > 1:
            print("--incremental is single-GPU (BASELINE configs[4])",
                  file=sys.stderr)
            sys.exit(2)
        import numpy as np
        from reth_amd.engine import DELTA_DTYPE, STORAGE_DTYPE
        nd = args.delta_accounts
        n_mod, n_del = (nd * 4) // 5, nd // 10
        n_new = nd - n_mod - n_del
        counters = list(range(n_mod))             + list(range(args.accounts, args.accounts + n_new))             + list(range(n_mod, n_mod + n_del))
        msgs = torch.zeros((len(counters), 8), dtype=torch.uint8,
                           device=f"cuda:{local_rank}")
        for b in range(8):
            msgs[:, b] = torch.tensor([(c >> (8 * b)) & 0xFF for c in counters],
                                      dtype=torch.uint8)
        dk = torch.empty((len(counters), 32), dtype=torch.uint8,
                         device=f"cuda:{local_rank}")
        eng.keccak_batch_device(msgs, 8, dk)
        dk = dk.cpu().numpy()
        ke = bytes.fromhex("c5d2460186f7233c927e7db2dcc703c0e500b653"
                           "ca82273b7bfad8045d85a470")
        rows = []
        for i, c in enumerate(counters):
            key = dk[i].tobytes()
            if i < n_mod:            # absolute-valued modification
                rows.append((key, c & 0xFFFF, 10**18 + c, ke, 0))
            elif i < n_mod + n_new:  # insert
                rows.append((key, 1, 5 * 10**17 + c, ke, 0))
            else:                    # delete
                rows.append((key, 0, 0, ke, 1))
        d = np.zeros(len(rows), dtype=DELTA_DTYPE)
        for i, (k, nn, b, ch, dead) in enumerate(sorted(rows)):
            d[i]["key"] = np.frombuffer(k, np.uint8)
            d[i]["nonce"] = nn
            d[i]["balance"] = np.frombuffer(b.to_bytes(32, "big"), np.uint8)
            d[i]["code_hash"] = np.frombuffer(ch, np.uint8)
            d[i]["deleted"] = dead
        st_rows = np.zeros(0, dtype=STORAGE_DTYPE)
        if args.incremental_slots > 0:
            # idempotent storage delta: absolute-valued upserts of 4 slots
            # per modified account (slots this delta itself introduces, so
            # the post-state is a fixed point after the first step)
            smsgs = torch.zeros((n_mod * 4, 12), dtype=torch.uint8,
                                device=f"cuda:{local_rank}")
            for b in range(8):
                smsgs[:, b] = torch.tensor(
                    [(c >> (8 * b)) & 0xFF for c in counters[:n_mod]
                     for _ in range(4)], dtype=torch.uint8)
            smsgs[:, 8] = torch.tensor([q for _ in range(n_mod)
                                        for q in range(4)], dtype=torch.uint8)
            smsgs[:, 9] = 0xD5  # domain tag: delta slots
            skeys = torch.empty((n_mod * 4, 32), dtype=torch.uint8,
                                device=f"cuda:{local_rank}")
            eng.keccak_batch_device(smsgs, 12, skeys)
            skeys = skeys.cpu().numpy()
            srows = []
            for i in range(n_mod):
                ak = dk[i].tobytes()
                for q in range(4):
                    srows.append((ak, skeys[4 * i + q].tobytes(),
                                  10**9 + 7 * i + q))
            srows.sort()
            st_rows = np.zeros(len(srows), dtype=STORAGE_DTYPE)
            for i, (ak, sk, v) in enumerate(srows):
                st_rows[i]["acct_key"] = np.frombuffer(ak, np.uint8)
                st_rows[i]["slot_key"] = np.frombuffer(sk, np.uint8)
                st_rows[i]["value"] = np.frombuffer(v.to_bytes(32, "big"),
                                                    np.uint8)
        delta = (d, st_rows)

    def step():
        if args.incremental and args.dirty:
            return eng.incremental_root(delta[0], delta[1])
        if args.incremental:
            eng.apply_delta(*delta)
            return eng.root()
        if world > 1:
            refs, lens, roots, counts = eng.subtree_roots()
            m = sharding.all_gather_combine(refs, lens, roots, counts,
                                            device=f"cuda:{local_rank}")
            return eng.finish_top(*m)
        return eng.root()

    def barrier():
        if dist:
            dist.barrier()
        torch.cuda.synchronize()

    # ---- warmup ----
    if args.incremental and args.dirty:
        eng.root_retaining()  # arm cell-top retention (untimed, once)
    root0 = None
    for wi in range(max(args.warmup, 1)):
        root0 = step()
        if wi == 0 and args.incremental:
            # the first delta replaced the resident state with the merged
            # (engine-owned) arrays; the borrowed generation tensors are
            # dead weight (62 GB at 10M x 64) — drop them
            eng.release_borrowed()
            acct_t = st_t = None  # noqa: F841
            torch.cuda.empty_cache()

    # ---- timed region ----
    barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        root = step()
        assert root == root0, "nondeterministic root across steps"
    barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=f"cuda:{local_rank}")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed * 1000.0 / args.steps
    stats = eng.stats()

    # optional property check (untimed): sharded composition equals root
    if args.check and world == 1:
        refs, lens, roots_, counts = eng.subtree_roots()
        assert eng.finish_top(refs, lens, roots_, counts) == root0, \
            "subtree composition mismatch"

    if rank != 0:
        if dist:
            dist.destroy_process_group()
        return

    # ---- roofline of the dominant kernel (leaf RLP+keccak), HIP-event timed
    # inside libsre on its own stream. Algorithmic bytes per leaf:
    # storage leaf 96 B (64 read + 32 ref write), account leaf 142 B.
    total_storage_leaves = ns if world == 1 else args.accounts * args.slots
    total_accounts = na if world == 1 else args.accounts
    # per-rank stats cover the rank's shard; at N=1 they are the whole job
    leaf_bytes = ns * 96 + na * 142
    leaf_s = stats["leaf_hash_ms"] / 1000.0
    achieved_gbps = leaf_bytes / leaf_s / 1e9 if leaf_s > 0 else 0.0
    keccak_ghs = stats["leaf_blocks"] / leaf_s / 1e9 if leaf_s > 0 else 0.0
    hash_blocks = stats["leaf_blocks"] + stats["branch_blocks"]
    hash_s = (stats["leaf_hash_ms"] + stats["branch_hash_ms"]) / 1000.0

    # ---- CPU baseline: the C-oracle restatement of reth's algorithm
    # (kind "port"), one thread per host core (okc_state_root_par).
    # Default ("full", non-incremental runs): the ACTUAL full-shape state —
    # the same device-generated entries copied D2H — measured unscaled,
    # with the oracle root asserted equal to the GPU root (a full-scale
    # parity pin for free). The single-thread leg stays sampled+scaled
    # (a full 1-core run is ~33 min) and is labeled as such.
    cpu_mode = "off" if args.no_cpu_baseline else args.cpu_baseline
    if args.incremental and cpu_mode == "full":
        cpu_mode = "sample"  # the incremental step has no full-shape CPU twin
    cpu_baseline = None
    cpu_baseline_1core = None
    if world == 1 and cpu_mode != "off":
        import os as _os
        import numpy as np
        from oracle import bind
        ncores = _os.cpu_count() or 1
        full_leaves = total_accounts + total_storage_leaves
        # single-thread leg: bounded sample, scaled (labeled)
        sa = min(args.cpu_sample_accounts, args.accounts)
        acct_s, st_s = gen.gen_state_numpy(sa, args.slots, bind.keccak256_batch)
        c0 = time.perf_counter()
        r1 = bind.state_root(acct_s, st_s)
        c1 = time.perf_counter()
        sample_leaves = len(acct_s) + len(st_s)
        scale = full_leaves / sample_leaves
        cpu_baseline_1core = {
            "value": round((c1 - c0) * 1000.0 * scale, 1),
            "unit": "ms",
            "cores": 1,
            "kind": "port",
            "sample": f"oracle (C) on {sa} accounts x {args.slots} slots = "
                      f"{sample_leaves} leaves, {(c1 - c0):.1f}s measured, "
                      f"scaled linearly in leaves to {full_leaves} leaves",
        }
        if cpu_mode == "full" and acct_t is not None:
            # threaded leg, full shape, unscaled: same entries D2H
            acct_h = np.ascontiguousarray(
                acct_t.cpu().numpy()).ravel().view(bind.ACCOUNT_DTYPE)
            st_h = np.ascontiguousarray(
                st_t.cpu().numpy()).ravel().view(bind.STORAGE_DTYPE)
            c1 = time.perf_counter()
            rp = bind.state_root_par(acct_h, st_h)
            c2 = time.perf_counter()
            assert rp == root0, "CPU oracle disagrees with GPU root at " \
                                "the full benchmark shape"
            cpu_baseline = {
                "value": round((c2 - c1) * 1000.0, 1),
                "unit": "ms",
                "cores": ncores,
                "kind": "port",
                "sample": f"full shape, measured: oracle (C, OpenMP, "
                          f"{ncores} threads) on the identical "
                          f"device-generated {full_leaves}-leaf state "
                          f"({(c2 - c1):.1f}s; oracle root == GPU root)",
            }
        else:
            rp = bind.state_root_par(acct_s, st_s)
            c2 = time.perf_counter()
            assert r1 == rp
            cpu_baseline = {
                "value": round((c2 - c1) * 1000.0 * scale, 1),
                "unit": "ms",
                "cores": ncores,
                "kind": "port",
                "sample": f"oracle (C, OpenMP, {ncores} threads) on "
                          f"{sa} accounts x {args.slots} slots = "
                          f"{sample_leaves} leaves, {(c2 - c1):.1f}s "
                          f"measured, scaled linearly in leaves to the "
                          f"full {full_leaves}-leaf job",
            }

    out = {
        "metric": "state-root wall-clock, 10M accounts x 64 slots",
        "value": round(ms_per_step, 3),
        "unit": "ms",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": False,
        "scaling": "strong",
        "vs_baseline": None,
        "dtype": "u8",
        "data": "synthetic (SURVEY.md §8d, seed 0x5EED, generated on-device)",
        "config": {
            "workload": (f"incremental: {args.accounts}-account resident "
                         f"base + {args.delta_accounts}-account overlay "
                         "delta per step (BASELINE configs[4]; accounts-only "
                         "base; " +
                         ("dirty-path recompute via retained cell tops"
                          if args.dirty else
                          "apply_delta + full device recompute") +
                         (f"; +{args.delta_accounts * 4 * 4 // 5} storage "
                          "delta rows" if args.incremental_slots else "") +
                         ")")
            if args.incremental else
            (f"{args.accounts} accounts x {args.slots} slots "
             "(BASELINE configs[3] shape; full job on every N)"),
            "accounts": args.accounts,
            "slots_per_account": args.slots,
            "storage_leaves": int(total_storage_leaves),
            "parallelism": f"top-nibble sharding x{world}, RCCL all-gather"
                           if world > 1 else "single GPU",
        },
        "roofline": {
            "bound": "hbm",
            "achieved": round(achieved_gbps, 1),
            "peak": 8000.0,
            "unit": "GB/s",
            "frac": round(achieved_gbps / 8000.0, 4),
            "traffic": _pmc_traffic(args.accounts, args.slots),
            "note": "leaf RLP+keccak kernel; algorithmic bytes "
                    "(96 B/storage leaf, 142 B/account leaf) / HIP-event "
                    "kernel time. The kernel is integer-VALU bound, not "
                    "HBM bound: chip Keccak-f ceiling ~13.6 GH/s at 78.6 "
                    "Tops/s u32 VALU peak (DESIGN.md §roofline).",
        } if not (args.incremental and args.dirty) else {
            "bound": "hbm", "achieved": None, "peak": 8000.0,
            "unit": "GB/s", "frac": None, "traffic": None,
            "note": "dirty-path incremental: the leaf kernel touches only "
                    "delta-dirty cells, so the full-job algorithmic-bytes "
                    "formula does not apply; dominant cost is the O(na) "
                    "merge/lcp/revalidate scans (see engine_stats and the "
                    "full-rebuild bench line for the kernel roofline).",
        },
        "cpu_baseline": cpu_baseline,
        "cpu_baseline_1core": cpu_baseline_1core,
        "keccak_ghs_leaf_kernel": round(keccak_ghs, 2),
        "keccak_ghs_all_kernels": round(hash_blocks / hash_s / 1e9, 2)
        if hash_s > 0 else 0.0,
        "engine_stats": {k: (round(v, 3) if isinstance(v, float) else int(v))
                         for k, v in stats.items()},
        "gen_seconds": round(t_gen, 1),
    }
    print(json.dumps(out))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
