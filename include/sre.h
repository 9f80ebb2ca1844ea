/* sre.h — C-ABI drop-in boundary of the MI355X-native state-root engine.
 *
 * This is the extern-C surface a reth maintainer would bind over FFI to
 * replace the Merkle-stage state-root computation. Each entry point cites the
 * reference interface it replaces (paths relative to /root/reference):
 *
 *   - sre_upload_accounts / sre_upload_storage + sre_root mirror
 *     `DatabaseStateRoot::overlay_root(tx, HashedPostStateSorted)`
 *     (crates/trie/db/src/state.rs:119-138) and the object-safe
 *     `StateRootProvider::state_root(HashedPostState) -> B256`
 *     (crates/storage/storage-api/src/trie.rs:13-41): sorted hashed entries
 *     in, 32-byte root out.
 *   - sre_storage_roots mirrors `StorageRootProvider::storage_root`
 *     (crates/storage/storage-api/src/trie.rs:45-58) /
 *     `StorageRoot::calculate_with_cursors` (crates/trie/trie/src/trie.rs:750-876)
 *     batched over every account.
 *   - sre_subtree_roots / sre_finish_top are the multi-GPU decomposition of
 *     `StateRoot::calculate`'s account-trie walk (crates/trie/trie/src/trie.rs:171-374):
 *     per-top-nibble subtrie digests exchanged over RCCL, root finished from
 *     the 16 child references.
 *
 * Input contract (matches the reference's cursor semantics):
 *   - account entries sorted ascending by `key` (the keccak256 of the
 *     address — reth's HashedAccounts table order,
 *     crates/trie/common/src/hashed_state.rs:331 `into_sorted`), no
 *     duplicates, no "empty" accounts (reth never stores them:
 *     crates/trie/trie/src/hashed_cursor/post_state.rs:70-82).
 *   - storage entries sorted ascending by (acct_key, slot_key), no
 *     duplicates, values nonzero (zero-valued slots are absent, not stored:
 *     crates/trie/trie/src/trie.rs:819-825), and every acct_key present in
 *     the accounts upload.
 *   - capacity: at most 2^32 - 1 account entries and 2^32 - 1 storage
 *     entries per context (leaf intervals are tracked as u32 indices).
 *     That is ~4.29 billion leaves per GPU — 6.6x the headline 650M-leaf
 *     config; shard across GPUs (sre_subtree_roots) beyond it. Exceeding
 *     the limit is rejected at upload with an error, never truncated.
 *
 * Thread model: all calls on one host thread per ctx; the engine is
 * internally multi-stream. The engine COPIES input buffers at upload; the
 * caller owns them afterwards. Errors: nonzero int return +
 * sre_last_error(). This library is GPU-only by design: sre_create fails
 * loudly when no HIP device is present — there is no CPU fallback.
 */
#ifndef SRE_H
#define SRE_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Hashed account entry — reth `Account` (external reth-primitives-traits) +
 * hashed address key; bytecode_hash None is represented as KECCAK_EMPTY, the
 * same normalization `into_trie_account` applies
 * (crates/stages/stages/src/stages/merkle.rs:297). balance is big-endian
 * 32-byte U256. 104 bytes, no padding. */
typedef struct {
    uint8_t  key[32];       /* keccak256(address) */
    uint64_t nonce;
    uint8_t  balance[32];   /* big-endian U256 */
    uint8_t  code_hash[32]; /* KECCAK_EMPTY when no code */
} sre_account_entry;

/* Hashed storage entry — one row of reth's HashedStorages dup-sorted table
 * (crates/storage/db-api/src/tables/mod.rs:481-507). value big-endian U256,
 * nonzero. 96 bytes, no padding. */
typedef struct {
    uint8_t acct_key[32];   /* keccak256(address) */
    uint8_t slot_key[32];   /* keccak256(slot) */
    uint8_t value[32];      /* big-endian U256, != 0 */
} sre_storage_entry;

typedef struct sre_ctx sre_ctx;

/* One stored trie node — the engine's representation of reth's
 * `BranchNodeCompact` + its path (crates/trie/common/src/updates.rs:17;
 * node shape per crates/trie/common/src/hash_builder/state.rs:15-48 and the
 * pinned expectations of crates/trie/db/tests/trie.rs:519-589 /
 * crates/trie/trie/src/node_iter.rs:383-465):
 *   - a branch node is stored iff hash_mask != 0;
 *   - hash_mask bit c is set iff child c's subtree top (after any extension)
 *     is a branch whose RLP is >= 32 bytes; `hashes` holds those branch-RLP
 *     keccaks in ascending nibble order;
 *   - tree_mask bit c is set iff child c's subtree top branch is itself
 *     stored;
 *   - the path-[] row (root is a branch) carries root_hash.
 * kind 0 = account trie; kind 1 = storage trie of acct_key. 624 bytes.
 *
 * `removed` (incremental mode only; always 0 from full rebuilds):
 *   0 = upsert (the row is stored at this path now);
 *   1 = removal — the path was stored before the delta and no longer is
 *       (reth TrieUpdates.removed_nodes / StorageTrieUpdates.removed_nodes,
 *       recorded by the walker at crates/trie/trie/src/walker.rs:363-369;
 *       masks/hashes fields are zero);
 *   2 = whole-storage-trie deletion marker for a destroyed account
 *       (kind 1, path_len 0) — reth's StorageTrieUpdates::set_deleted(true)
 *       for prefix_sets.destroyed_accounts
 *       (crates/trie/common/src/updates.rs:154-157). */
typedef struct {
    uint8_t  acct_key[32];   /* kind 1 only */
    uint8_t  kind;
    uint8_t  path_len;       /* nibbles, 0..63 */
    uint8_t  path[32];       /* packed nibbles, high nibble first */
    uint8_t  num_hashes;     /* popcount(hash_mask) */
    uint8_t  root_hash_set;  /* 1 => root_hash valid (path_len == 0) */
    uint16_t state_mask, tree_mask, hash_mask;
    uint8_t  root_hash[32];
    uint8_t  hashes[16][32];
    uint8_t  removed;        /* see above */
    uint8_t  pad_[5];
} sre_update_row;

/* Per-call timing/throughput stats (HIP-event measured, for bench reporting;
 * mirrors the spirit of crates/trie/trie/src/metrics.rs:7-60). Times in ms. */
typedef struct {
    double   total_ms;          /* whole sre_root device pipeline */
    double   leaf_hash_ms;      /* storage+account leaf RLP+keccak kernels */
    uint64_t leaf_count;        /* leaves hashed (storage + account) */
    uint64_t leaf_blocks;       /* keccak-f[1600] invocations in leaf kernels */
    double   branch_hash_ms;    /* branch/extension assemble+keccak kernels */
    uint64_t branch_count;      /* branch nodes built */
    uint64_t branch_blocks;     /* keccak-f invocations in branch kernels */
    double   sort_ms;           /* lcp/bucket/merge machinery */
    uint64_t levels;            /* depth levels processed */
} sre_stats;

/* Create a context bound to HIP device `device`. Returns NULL on failure
 * (use sre_last_error(NULL) for the reason). */
sre_ctx *sre_create(int device);
void     sre_destroy(sre_ctx *ctx);

/* Upload sorted entries to device memory (H2D copy happens here, outside any
 * timed region). May be called again to replace the state. */
int sre_upload_accounts(sre_ctx *ctx, const sre_account_entry *entries, uint64_t n);
int sre_upload_storage(sre_ctx *ctx, const sre_storage_entry *entries, uint64_t n);

/* Compute the state root over the uploaded entries.
 * Equivalent surface: StateRoot::root() (crates/trie/trie/src/trie.rs:154). */
int sre_root(sre_ctx *ctx, uint8_t out_root[32]);

/* Per-account storage roots (n must equal the uploaded account count; out is
 * n*32 bytes, account order). Runs the storage pass only. */
int sre_storage_roots(sre_ctx *ctx, uint8_t *out, uint64_t n);

/* Shard mode: compute, for each account top nibble owned by this rank's
 * upload, (a) the child reference the root branch would embed (len in
 * out_child_lens[i]; 0 = nibble absent) and (b) the 32-byte root hash the
 * subtree would have if it were the entire trie, plus leaf counts. The
 * caller all-gathers these across ranks and calls sre_finish_top. */
int sre_subtree_roots(sre_ctx *ctx,
                      uint8_t  out_child_refs[16][33],
                      uint8_t  out_child_lens[16],
                      uint8_t  out_root_hash[16][32],
                      uint64_t out_counts[16]);

/* Finish the account-trie top from gathered per-nibble results.
 * counts[i] = total accounts under nibble i across all ranks. */
int sre_finish_top(sre_ctx *ctx,
                   const uint8_t  child_refs[16][33],
                   const uint8_t  child_lens[16],
                   const uint8_t  root_hash[16][32],
                   const uint64_t counts[16],
                   uint8_t out_root[32]);

/* One account row of a HashedPostState overlay delta. deleted=1 destroys
 * the account AND wipes its storage (HashedStorage::new(wiped=true),
 * crates/trie/common/src/hashed_state.rs:425-440). 112 bytes. */
typedef struct {
    uint8_t  key[32];
    uint64_t nonce;
    uint8_t  balance[32];
    uint8_t  code_hash[32];
    uint8_t  deleted;
    uint8_t  pad_[7];
} sre_account_delta;

/* Apply a HashedPostState delta to the resident state — the overlay-merge
 * semantics of HashedPostStateCursor (crates/trie/trie/src/hashed_cursor/
 * post_state.rs:89,313,355): post-state wins, a zero storage value deletes
 * the slot, deleted accounts disappear along with all their storage. Both
 * delta arrays sorted ascending (by key / (acct_key, slot_key)). The
 * resident state is REPLACED by the merged result (device-resident), so a
 * following sre_root computes the post-delta root — the incremental-root
 * entry (DatabaseStateRoot::incremental_root semantics at the state level,
 * crates/trie/db/src/state.rs:62; BASELINE configs[4]). A storage delta row
 * for an account deleted in the same delta is invalid. */
int sre_apply_delta(sre_ctx *ctx,
                    const sre_account_delta *acct_delta, uint64_t n_acct,
                    const sre_storage_entry *st_delta, uint64_t n_st);

/* Dirty-path incremental root.
 *
 * sre_root_retaining computes the full root like sre_root AND retains (a)
 * the engine's cell-top records (the unique trie nodes covering fixed
 * 5-nibble account-key prefix cells) and (b) every account's storage
 * root. sre_incremental_root then applies an overlay delta (semantics of
 * sre_apply_delta: account upserts/deletes + storage upserts/zero-deletes)
 * and recomputes ONLY what the delta touches: the storage tries of
 * accounts with storage-delta rows (other accounts' roots are carried
 * across the merge; destroyed accounts drop theirs), and the account-trie
 * cells containing any delta key — seeding every clean cell's retained
 * top record into the level machinery. This is the walker-skip semantics
 * of StateRoot::root with a prefix set (crates/trie/trie/src/trie.rs:228-253
 * `walker.advance` over `changed_prefixes`), expressed as cell-granular
 * reuse. The resident state is replaced by the merged result and retention
 * is refreshed, so deltas chain. Uncovered positions fall back to leaf
 * recomputation (still correct). */
int sre_root_retaining(sre_ctx *ctx, uint8_t out_root[32]);
int sre_incremental_root(sre_ctx *ctx,
                         const sre_account_delta *acct_delta,
                         uint64_t n_acct,
                         const sre_storage_entry *st_delta,
                         uint64_t n_st, uint8_t out_root[32]);

/* Incremental root WITH TrieUpdates — the engine's equivalent of
 * StateRoot::root_with_updates driven by a prefix set, i.e. the Merkle
 * stage's incremental regime (crates/stages/stages/src/stages/merkle.rs:
 * 319-342 -> incremental_root_with_updates, crates/trie/db/src/state.rs:62),
 * including removed_nodes (walker.rs:363-369) and destroyed-account
 * storage-trie deletion (updates.rs:140,154-157).
 *
 * Arm with sre_root_retaining_with_updates: computes the full root,
 * retains cell-tops AND the full stored-row set (readable via
 * sre_updates_count/get, all rows removed=0). Each following
 * sre_incremental_root_with_updates applies the delta, recomputes dirty
 * paths, and exposes the NET row diff vs the pre-delta trie:
 * upserts (removed=0, new or changed rows), removals (removed=1), and
 * destroyed-account markers (removed=2). Applying the diff to the
 * pre-delta row set yields exactly the full-rebuild row set of the
 * post-delta state — the same database effect as reth's
 * finalize + into_sorted stream (which may additionally re-write
 * unchanged rows on dirty paths; the engine suppresses those).
 * Rows are ordered like TrieUpdatesSorted (account rows path-sorted
 * first, upserts and removals interleaved; then storage rows grouped by
 * account). Deltas chain; a plain sre_incremental_root or any state
 * replacement disarms the row retention (rearm with
 * sre_root_retaining_with_updates). */
int sre_root_retaining_with_updates(sre_ctx *ctx, uint8_t out_root[32]);
int sre_incremental_root_with_updates(sre_ctx *ctx,
                                      const sre_account_delta *acct_delta,
                                      uint64_t n_acct,
                                      const sre_storage_entry *st_delta,
                                      uint64_t n_st, uint8_t out_root[32]);

/* state_root_from_nodes equivalent — the remaining StateRootProvider
 * methods (state_root_from_nodes{,_with_updates} over TrieInput,
 * crates/storage/storage-api/src/trie.rs:26-40; TrieInput
 * crates/trie/common/src/input.rs:10): compute the root of
 * (resident state + delta) using `rows` — a stored-node overlay in the
 * engine's own sre_update_row format, e.g. the output of
 * sre_root_with_updates — in place of recomputing every storage trie.
 * kind-1 path-[] rows' root_hash short-circuits untouched accounts'
 * storage roots exactly like reth's walker skip over stored roots
 * (walker.rs:195-230); tries without a usable row or touched by the
 * delta are rebuilt from their entries; the account trie is rebuilt
 * on-device (it is ~2% of the leaf work — the account-node skip that
 * kind-0 rows enable on a CPU walk saves nothing here, so those rows
 * are accepted and ignored). Removal rows (removed != 0) are invalid
 * input. The resident state is REPLACED by the merged result. For the
 * _with_updates form, call sre_root_with_updates afterwards or use the
 * incremental surface above. */
int sre_root_from_nodes(sre_ctx *ctx,
                        const sre_update_row *rows, uint64_t n_rows,
                        const sre_account_delta *acct_delta, uint64_t n_acct,
                        const sre_storage_entry *st_delta, uint64_t n_st,
                        uint8_t out_root[32]);

/* Account multiproof — the surface of Proof::account_proof /
 * Proof::multiproof restricted to account targets
 * (crates/trie/trie/src/proof/mod.rs:59-137 `multiproof`, collecting the
 * RLP of every node on each target's path root-first, the ProofRetainer
 * semantics of the alloy-trie HashBuilder at trie.rs:292-300). Present
 * targets yield the path to their leaf; ABSENT targets yield the
 * exclusion proof — the lookup-path nodes ending at the proven divergence
 * (empty branch slot or mismatching extension/leaf path); an empty state
 * yields an empty list. Nodes are returned
 * root-first per target, concatenated: out_nodes holds the RLP bytes
 * back-to-back, out_lens one length per node, out_counts one node count
 * per target (in target order). cap_* are capacities in bytes / entries;
 * fails loudly when exceeded. Account-trie proofs never contain inline
 * (<32 B) nodes: every account leaf RLP is >= 70 B, so every referenced
 * node on the path is hashed. */
int sre_account_proof(sre_ctx *ctx,
                      const uint8_t *targets /* n x 32, hashed keys */,
                      uint64_t n_targets,
                      uint8_t *out_nodes, uint64_t cap_nodes,
                      uint32_t *out_lens, uint64_t cap_lens,
                      uint32_t *out_counts);

/* Storage multiproof — StorageProof::storage_multiproof
 * (crates/trie/trie/src/proof/mod.rs) for (acct_key, slot_key)
 * pairs. The slot may be absent (exclusion semantics as in
 * sre_account_proof); a storage-less OR ABSENT account yields
 * EMPTY_ROOT_HASH and an empty node list — exactly
 * StorageMultiProof::empty(), the reference's short-circuit on an empty
 * storage cursor (proof/mod.rs storage_multiproof): per target the account's storage
 * root (out_roots, 32 B each) and the root-first node list of its storage
 * trie, same output layout as sre_account_proof. Storage tries CAN
 * contain inline (<32 B) nodes; those are embedded in their parents and
 * not emitted. */
int sre_storage_proof(sre_ctx *ctx,
                      const uint8_t *acct_keys /* n x 32 */,
                      const uint8_t *slot_keys /* n x 32 */,
                      uint64_t n_targets, uint8_t *out_roots,
                      uint8_t *out_nodes, uint64_t cap_nodes,
                      uint32_t *out_lens, uint64_t cap_lens,
                      uint32_t *out_counts);

/* Compute the state root AND retain the stored trie nodes (TrieUpdates) —
 * the surface of StateRootProvider::state_root_with_updates
 * (crates/storage/storage-api/src/trie.rs:30) / StateRoot::root_with_updates
 * feeding TrieWriter::write_trie_updates (storage-api/src/trie.rs:176-188).
 * Rows are retrieved afterwards with sre_updates_count / sre_updates_get
 * (account-trie rows first, then storage rows grouped by account; each list
 * sorted by path). A full rebuild has no removed_nodes. */
int sre_root_with_updates(sre_ctx *ctx, uint8_t out_root[32]);
int64_t sre_updates_count(sre_ctx *ctx);
int sre_updates_get(sre_ctx *ctx, sre_update_row *out, uint64_t max_rows);

int sre_get_stats(sre_ctx *ctx, sre_stats *out);

/* Last error string for ctx (or the global creation error when ctx==NULL). */
const char *sre_last_error(const sre_ctx *ctx);

#ifdef __cplusplus
}
#endif
#endif /* SRE_H */
