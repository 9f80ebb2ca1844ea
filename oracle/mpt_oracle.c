/* mpt_oracle.c — CPU restatement of reth's state-root computation.
 *
 * TEST INFRASTRUCTURE + REPORTED CPU BASELINE ONLY. Only oracle/, tests/,
 * __graft_entry__.smoke() and bench.py's cpu_baseline leg may call this; the
 * product path is the HIP engine (reth_amd/csrc) and must fail loudly rather
 * than fall back here.
 *
 * Restates (paths relative to /root/reference):
 *   - StateRoot::calculate account walk + per-leaf storage roots:
 *     crates/trie/trie/src/trie.rs:171-374 (hot loop :276-351)
 *   - StorageRoot::calculate_with_cursors inner loop: trie.rs:750-876
 *     (empty-storage shortcut -> EMPTY_ROOT_HASH :771-781)
 *   - storage leaf value = encode_fixed_size(U256): trie.rs:819-825,
 *     proof_v2/value.rs:55-66 (minimal big-endian RLP)
 *   - account leaf value = RLP([nonce, balance, storage_root, code_hash]),
 *     <= 110 B: trie.rs:472-476, crates/trie/common/src/root.rs:9-31,
 *     proof_v2/value.rs:125-139
 *   - node -> reference rule (RLP if <32 B else 0xa0||keccak; root always
 *     hashed): proof_v2/node.rs:40-66 RlpNode::from_rlp usage,
 *     proof_v2/mod.rs:1628 compute_root_hash
 *   - branch = 17-item list, extension = [HP(shared), child], hex-prefix
 *     encoding: spec-only in the reference (alloy-trie/nybbles, external);
 *     restated from the Yellow Paper; shapes confirmed by the expectations in
 *     crates/trie/trie/src/node_iter.rs:420-449.
 *
 * Algorithm: recursion over the sorted leaf stream by nibble partition —
 * deliberately different from both oracle/pyref.py (dict recursion) and the
 * HIP engine (bottom-up LCP levels) so the three implementations only agree
 * if the semantics agree. Parity pins: tests/golden (consensus genesis
 * roots + trie.rs:489 fixed vector) and random cross-checks vs pyref.
 */
#include <stdlib.h>
#include <string.h>

#include "../include/sre.h"

void okc_keccak256(const uint8_t *in, size_t len, uint8_t out[32]);

/* keccak(rlp("")) and keccak("") — pinned by tests against the reference's
 * EMPTY_ROOT_HASH / KECCAK_EMPTY usage (trie.rs:15,775; merkle.rs:297). */
static const uint8_t EMPTY_ROOT[32] = {
    0x56, 0xe8, 0x1f, 0x17, 0x1b, 0xcc, 0x55, 0xa6, 0xff, 0x83, 0x45, 0xe6,
    0x92, 0xc0, 0xf8, 0x6e, 0x5b, 0x48, 0xe0, 0x1b, 0x99, 0x6c, 0xad, 0xc0,
    0x01, 0x62, 0x2f, 0xb5, 0xe3, 0x63, 0xb4, 0x21,
};

typedef struct {
    uint8_t len;      /* bytes valid in b; ref as embedded in parent payload */
    uint8_t b[33];    /* 0xa0||hash (33) or inline RLP (<32) */
} ref_t;

/* ---------------- RLP helpers ---------------- */

static size_t rlp_str(const uint8_t *s, size_t len, uint8_t *out)
{
    if (len == 1 && s[0] < 0x80) {
        out[0] = s[0];
        return 1;
    }
    if (len < 56) {
        out[0] = (uint8_t)(0x80 + len);
        memcpy(out + 1, s, len);
        return 1 + len;
    }
    /* longest string here is 529-byte branch payloads' container, never a
     * string; leaf values <= 110. Two-byte length covers everything. */
    out[0] = 0xb8 + (len > 255 ? 1 : 0);
    if (len > 255) {
        out[1] = (uint8_t)(len >> 8);
        out[2] = (uint8_t)len;
        memcpy(out + 3, s, len);
        return 3 + len;
    }
    out[1] = (uint8_t)len;
    memcpy(out + 2, s, len);
    return 2 + len;
}

static size_t rlp_list_hdr(size_t payload_len, uint8_t *out)
{
    if (payload_len < 56) {
        out[0] = (uint8_t)(0xc0 + payload_len);
        return 1;
    }
    if (payload_len <= 255) {
        out[0] = 0xf8;
        out[1] = (uint8_t)payload_len;
        return 2;
    }
    out[0] = 0xf9;
    out[1] = (uint8_t)(payload_len >> 8);
    out[2] = (uint8_t)payload_len;
    return 3;
}

/* minimal big-endian integer bytes -> RLP scalar (0 -> 0x80) */
static size_t rlp_uint_be(const uint8_t *be, size_t width, uint8_t *out)
{
    size_t i = 0;
    while (i < width && be[i] == 0)
        i++;
    if (i == width) {
        out[0] = 0x80;
        return 1;
    }
    return rlp_str(be + i, width - i, out);
}

static size_t rlp_u64(uint64_t v, uint8_t *out)
{
    uint8_t be[8];
    for (int i = 0; i < 8; i++)
        be[i] = (uint8_t)(v >> (56 - 8 * i));
    return rlp_uint_be(be, 8, out);
}

/* hex-prefix encode nibbles key[from..to) of a 32-byte key */
static size_t hp_encode(const uint8_t *key, int from, int to, int leaf, uint8_t *out)
{
    int n = to - from;
    int odd = n & 1;
    size_t w = 0;
    out[w++] = (uint8_t)((leaf ? 0x20 : 0x00) | (odd ? 0x10 : 0x00));
    int i = from;
    if (odd) {
        out[0] |= (uint8_t)((key[i / 2] >> ((i & 1) ? 0 : 4)) & 0x0f);
        i++;
    }
    for (; i < to; i += 2) {
        uint8_t hi = (key[i / 2] >> ((i & 1) ? 0 : 4)) & 0x0f;
        uint8_t lo = (key[(i + 1) / 2] >> (((i + 1) & 1) ? 0 : 4)) & 0x0f;
        out[w++] = (uint8_t)((hi << 4) | lo);
    }
    return w;
}

static uint8_t nib(const uint8_t *key, int i)
{
    return (key[i / 2] >> ((i & 1) ? 0 : 4)) & 0x0f;
}

static void make_ref(const uint8_t *rlp, size_t len, ref_t *ref)
{
    if (len < 32) {
        ref->len = (uint8_t)len;
        memcpy(ref->b, rlp, len);
    } else {
        ref->len = 33;
        ref->b[0] = 0xa0;
        okc_keccak256(rlp, len, ref->b + 1);
    }
}

/* ---------------- generic sorted-stream trie build ---------------- */

/* TrieUpdates collector (see sre_update_row in sre.h for the semantics,
 * pinned by /root/reference/crates/trie/db/tests/trie.rs:519-589 and
 * crates/trie/trie/src/node_iter.rs:383-465). */
typedef struct {
    sre_update_row *rows;
    uint64_t n, cap;
    uint8_t kind;                /* 0 account trie, 1 storage trie */
    const uint8_t *acct_key;     /* kind 1 */
} upd_ctx;

/* what a parent branch needs to know about a child subtree */
typedef struct {
    uint8_t is_branch;           /* subtree top (after ext) is a branch */
    uint8_t branch_hashed;       /* that branch's RLP >= 32 bytes */
    uint8_t stored;              /* that branch was emitted (hash_mask != 0) */
    uint8_t branch_hash[32];
} child_info;

typedef struct {
    const uint8_t *keys;   /* 32-byte keys at key_stride apart, sorted asc */
    size_t key_stride;
    /* writes the leaf's RLP-encoded value for leaf idx, returns length */
    size_t (*value)(const void *vctx, uint64_t idx, uint8_t *out);
    const void *vctx;
    upd_ctx *upd;          /* optional TrieUpdates collector */
} trie_src;

static int upd_push(upd_ctx *u, const sre_update_row *row)
{
    if (u->n == u->cap) {
        u->cap = u->cap ? u->cap * 2 : 64;
        u->rows = (sre_update_row *)realloc(u->rows, u->cap * sizeof(*u->rows));
        if (!u->rows)
            return -1;
    }
    u->rows[u->n++] = *row;
    return 0;
}

#define NODE_MAX 532 /* 17-item branch: 16*33+1 payload + 3-byte header */

/* Builds the node covering leaves [lo,hi) whose keys all agree on nibbles
 * [0,pos). Writes the node's RLP to out (caller buffer >= NODE_MAX), returns
 * its length, and fills *ref with the embedded reference. */
static size_t build(const trie_src *src, uint64_t lo, uint64_t hi, int pos,
                    uint8_t *out, ref_t *ref, child_info *ci)
{
    if (ci)
        memset(ci, 0, sizeof(*ci));
    const uint8_t *key_lo = src->keys + lo * src->key_stride;
    if (hi - lo == 1) {
        uint8_t val[128];
        size_t val_len = src->value(src->vctx, lo, val);
        uint8_t hp[33];
        size_t hp_len = hp_encode(key_lo, pos, 64, 1, hp);
        uint8_t payload[NODE_MAX];
        size_t p = rlp_str(hp, hp_len, payload);
        p += rlp_str(val, val_len, payload + p);
        size_t h = rlp_list_hdr(p, out);
        memcpy(out + h, payload, p);
        make_ref(out, h + p, ref);
        return h + p;
    }
    /* common prefix of the sorted range = lcp(first, last) */
    const uint8_t *key_hi = src->keys + (hi - 1) * src->key_stride;
    int p = pos;
    while (p < 64 && nib(key_lo, p) == nib(key_hi, p))
        p++;
    if (p > pos) {
        /* extension node over key[pos..p) */
        ref_t child;
        uint8_t child_rlp[NODE_MAX];
        /* masks/storage see through the extension: the stored-trie model
         * keys nodes by the BRANCH path and reconstructs extensions */
        build(src, lo, hi, p, child_rlp, &child, ci);
        uint8_t hp[33];
        size_t hp_len = hp_encode(key_lo, pos, p, 0, hp);
        uint8_t payload[72];
        size_t w = rlp_str(hp, hp_len, payload);
        memcpy(payload + w, child.b, child.len);
        w += child.len;
        size_t h = rlp_list_hdr(w, out);
        memcpy(out + h, payload, w);
        make_ref(out, h + w, ref);
        return h + w;
    }
    /* branch at pos */
    uint8_t payload[NODE_MAX];
    sre_update_row row;
    memset(&row, 0, sizeof(row));
    int nhash = 0;
    size_t w = 0;
    uint64_t i = lo;
    for (int b = 0; b < 16; b++) {
        uint64_t j = i;
        while (j < hi && nib(src->keys + j * src->key_stride, pos) == b)
            j++;
        if (j == i) {
            payload[w++] = 0x80;
        } else {
            ref_t child;
            uint8_t child_rlp[NODE_MAX];
            child_info cci;
            build(src, i, j, pos + 1, child_rlp, &child, &cci);
            memcpy(payload + w, child.b, child.len);
            w += child.len;
            row.state_mask |= (uint16_t)(1u << b);
            if (cci.is_branch && cci.branch_hashed) {
                row.hash_mask |= (uint16_t)(1u << b);
                memcpy(row.hashes[nhash++], cci.branch_hash, 32);
            }
            if (cci.stored)
                row.tree_mask |= (uint16_t)(1u << b);
            i = j;
        }
    }
    payload[w++] = 0x80; /* value slot: always empty in state/storage tries */
    size_t h = rlp_list_hdr(w, out);
    memcpy(out + h, payload, w);
    make_ref(out, h + w, ref);
    if (ci || (src->upd && row.hash_mask)) {
        uint8_t bh[32];
        okc_keccak256(out, h + w, bh);
        if (ci) {
            ci->is_branch = 1;
            ci->branch_hashed = (h + w) >= 32;
            ci->stored = row.hash_mask != 0;
            memcpy(ci->branch_hash, bh, 32);
        }
        if (src->upd && row.hash_mask) { /* stored iff hash_mask != 0 */
            row.kind = src->upd->kind;
            if (src->upd->acct_key)
                memcpy(row.acct_key, src->upd->acct_key, 32);
            row.path_len = (uint8_t)pos;
            for (int k = 0; k < pos; k++) {
                uint8_t nb_ = nib(key_lo, k);
                if (k & 1)
                    row.path[k / 2] |= nb_;
                else
                    row.path[k / 2] = (uint8_t)(nb_ << 4);
            }
            row.num_hashes = (uint8_t)nhash;
            if (pos == 0) { /* root branch row carries the root hash */
                row.root_hash_set = 1;
                memcpy(row.root_hash, bh, 32);
            }
            upd_push(src->upd, &row);
        }
    }
    return h + w;
}

/* ---------------- leaf value encoders ---------------- */

static size_t storage_value(const void *vctx, uint64_t idx, uint8_t *out)
{
    const sre_storage_entry *e = (const sre_storage_entry *)vctx + idx;
    return rlp_uint_be(e->value, 32, out);
}

typedef struct {
    const sre_account_entry *accts;
    const uint8_t *storage_roots; /* 32 B per account */
} acct_vctx;

static size_t account_value(const void *vctx_, uint64_t idx, uint8_t *out)
{
    const acct_vctx *c = (const acct_vctx *)vctx_;
    const sre_account_entry *a = c->accts + idx;
    uint8_t payload[110];
    size_t w = rlp_u64(a->nonce, payload);
    w += rlp_uint_be(a->balance, 32, payload + w);
    w += rlp_str(c->storage_roots + 32 * idx, 32, payload + w);
    w += rlp_str(a->code_hash, 32, payload + w);
    size_t h = rlp_list_hdr(w, out);
    memmove(out + h, payload, w);
    return h + w;
}

/* ---------------- input validation ---------------- */

const char *okc_error_str(int code)
{
    switch (code) {
    case 0: return "ok";
    case 1: return "accounts not strictly ascending by key";
    case 2: return "storage not strictly ascending by (acct_key, slot_key)";
    case 3: return "storage entry for account not in accounts upload";
    case 4: return "zero-valued storage entry (zero slots must be absent)";
    case 5: return "bad arguments";
    default: return "unknown error";
    }
}

static int is_zero32(const uint8_t *p)
{
    for (int i = 0; i < 32; i++)
        if (p[i])
            return 0;
    return 1;
}

/* ---------------- public API ---------------- */

/* Per-account storage roots; also validates all input contracts.
 * roots_out: na*32 bytes. */
static int storage_roots_impl(const sre_account_entry *acct, uint64_t na,
                              const sre_storage_entry *st, uint64_t ns,
                              uint8_t *roots_out, upd_ctx *upd)
{
    for (uint64_t i = 1; i < na; i++)
        if (memcmp(acct[i - 1].key, acct[i].key, 32) >= 0)
            return 1;
    for (uint64_t i = 1; i < ns; i++) {
        int c = memcmp(st[i - 1].acct_key, st[i].acct_key, 32);
        if (c > 0 || (c == 0 && memcmp(st[i - 1].slot_key, st[i].slot_key, 32) >= 0))
            return 2;
    }
    for (uint64_t i = 0; i < ns; i++)
        if (is_zero32(st[i].value))
            return 4;

    trie_src src;
    src.upd = NULL;
    src.key_stride = sizeof(sre_storage_entry);
    src.value = storage_value;
    src.vctx = st;

    uint64_t i = 0;
    for (uint64_t j = 0; j < na; j++) {
        /* storage for accounts ordered like accounts: lockstep walk */
        if (i < ns && memcmp(st[i].acct_key, acct[j].key, 32) < 0)
            return 3; /* storage acct_key smaller than every remaining account */
        uint64_t lo = i;
        while (i < ns && memcmp(st[i].acct_key, acct[j].key, 32) == 0)
            i++;
        if (i == lo) {
            memcpy(roots_out + 32 * j, EMPTY_ROOT, 32); /* trie.rs:771-781 */
        } else {
            /* re-base keys AND value ctx so build's idx 0 == entry lo */
            src.keys = st[lo].slot_key;
            src.vctx = st + lo;
            if (upd) {
                upd->kind = 1;
                upd->acct_key = acct[j].key;
                src.upd = upd;
            }
            uint8_t rlp[NODE_MAX];
            ref_t ref;
            size_t len = build(&src, 0, i - lo, 0, rlp, &ref, NULL);
            okc_keccak256(rlp, len, roots_out + 32 * j);
        }
    }
    if (i != ns)
        return 3;
    return 0;
}

int okc_storage_roots(const sre_account_entry *acct, uint64_t na,
                      const sre_storage_entry *st, uint64_t ns,
                      uint8_t *roots_out)
{
    return storage_roots_impl(acct, na, st, ns, roots_out, NULL);
}

/* sort rows: account-trie rows first (by path), then storage rows grouped
 * by account key (by path within) — the order updates.rs `into_sorted`
 * produces. */
static int cmp_rows(const void *a_, const void *b_)
{
    const sre_update_row *a = (const sre_update_row *)a_;
    const sre_update_row *b = (const sre_update_row *)b_;
    if (a->kind != b->kind)
        return a->kind < b->kind ? -1 : 1;
    if (a->kind == 1) {
        int c = memcmp(a->acct_key, b->acct_key, 32);
        if (c)
            return c;
    }
    int minl = a->path_len < b->path_len ? a->path_len : b->path_len;
    for (int k = 0; k < minl; k++) {
        uint8_t na_ = (a->path[k / 2] >> ((k & 1) ? 0 : 4)) & 0xf;
        uint8_t nb_ = (b->path[k / 2] >> ((k & 1) ? 0 : 4)) & 0xf;
        if (na_ != nb_)
            return na_ < nb_ ? -1 : 1;
    }
    if (a->path_len != b->path_len)
        return a->path_len < b->path_len ? -1 : 1;
    return 0;
}

/* state root + TrieUpdates (StateRoot::root_with_updates semantics,
 * crates/trie/trie/src/trie.rs:141; full rebuild => no removed nodes). */
int okc_state_root_with_updates(const sre_account_entry *acct, uint64_t na,
                                const sre_storage_entry *st, uint64_t ns,
                                uint8_t out_root[32],
                                sre_update_row **rows_out, uint64_t *n_rows)
{
    *rows_out = NULL;
    *n_rows = 0;
    if (na == 0) {
        if (ns != 0)
            return 3;
        memcpy(out_root, EMPTY_ROOT, 32);
        return 0;
    }
    uint8_t *roots = (uint8_t *)malloc(na * 32);
    if (!roots)
        return 5;
    upd_ctx u;
    memset(&u, 0, sizeof(u));
    int rc = storage_roots_impl(acct, na, st, ns, roots, &u);
    if (rc) {
        free(roots);
        free(u.rows);
        return rc;
    }
    u.kind = 0;
    u.acct_key = NULL;
    acct_vctx vc = { acct, roots };
    trie_src src;
    src.upd = &u;
    src.keys = acct[0].key;
    src.key_stride = sizeof(sre_account_entry);
    src.value = account_value;
    src.vctx = &vc;
    uint8_t rlp[NODE_MAX];
    ref_t ref;
    size_t len = build(&src, 0, na, 0, rlp, &ref, NULL);
    okc_keccak256(rlp, len, out_root);
    free(roots);
    qsort(u.rows, u.n, sizeof(sre_update_row), cmp_rows);
    *rows_out = u.rows;
    *n_rows = u.n;
    return 0;
}

void okc_free_updates(sre_update_row *rows)
{
    free(rows);
}

int okc_state_root(const sre_account_entry *acct, uint64_t na,
                   const sre_storage_entry *st, uint64_t ns,
                   uint8_t out[32])
{
    if (na == 0) {
        if (ns != 0)
            return 3;
        memcpy(out, EMPTY_ROOT, 32);
        return 0;
    }
    uint8_t *roots = (uint8_t *)malloc(na * 32);
    if (!roots)
        return 5;
    int rc = okc_storage_roots(acct, na, st, ns, roots);
    if (rc) {
        free(roots);
        return rc;
    }
    acct_vctx vc = { acct, roots };
    trie_src src;
    src.upd = NULL;
    src.keys = acct[0].key;
    src.key_stride = sizeof(sre_account_entry);
    src.value = account_value;
    src.vctx = &vc;
    uint8_t rlp[NODE_MAX];
    ref_t ref;
    size_t len = build(&src, 0, na, 0, rlp, &ref, NULL);
    okc_keccak256(rlp, len, out);
    free(roots);
    return 0;
}

/* Shard mode: per-top-nibble child refs + standalone root hashes + counts.
 * Mirrors the decomposition in SURVEY.md §8(e). */
int okc_subtree_roots(const sre_account_entry *acct, uint64_t na,
                      const sre_storage_entry *st, uint64_t ns,
                      uint8_t out_child_refs[16][33], uint8_t out_child_lens[16],
                      uint8_t out_root_hash[16][32], uint64_t out_counts[16])
{
    memset(out_child_lens, 0, 16);
    memset(out_counts, 0, 16 * sizeof(uint64_t));
    if (na == 0)
        return ns == 0 ? 0 : 3;
    uint8_t *roots = (uint8_t *)malloc(na * 32);
    if (!roots)
        return 5;
    int rc = okc_storage_roots(acct, na, st, ns, roots);
    if (rc) {
        free(roots);
        return rc;
    }
    uint64_t i = 0;
    while (i < na) {
        uint8_t b = nib(acct[i].key, 0);
        uint64_t j = i;
        while (j < na && nib(acct[j].key, 0) == b)
            j++;
        acct_vctx vc = { acct + i, roots + 32 * i };
        trie_src src;
        src.upd = NULL;
        src.keys = acct[i].key;
        src.key_stride = sizeof(sre_account_entry);
        src.value = account_value;
        src.vctx = &vc;
        uint8_t rlp[NODE_MAX];
        ref_t ref;
        /* child-of-root-branch form: path consumed through nibble 0 */
        build(&src, 0, j - i, 1, rlp, &ref, NULL);
        memcpy(out_child_refs[b], ref.b, ref.len);
        out_child_lens[b] = ref.len;
        /* standalone-trie form (only used when this nibble is the whole trie) */
        size_t len = build(&src, 0, j - i, 0, rlp, &ref, NULL);
        okc_keccak256(rlp, len, out_root_hash[b]);
        out_counts[b] = j - i;
        i = j;
    }
    free(roots);
    return 0;
}

int okc_finish_top(const uint8_t child_refs[16][33], const uint8_t child_lens[16],
                   const uint8_t root_hash[16][32], const uint64_t counts[16],
                   uint8_t out[32])
{
    int populated = 0, last = -1;
    for (int b = 0; b < 16; b++)
        if (child_lens[b]) {
            populated++;
            last = b;
        }
    (void)counts;
    if (populated == 0) {
        memcpy(out, EMPTY_ROOT, 32);
        return 0;
    }
    if (populated == 1) {
        memcpy(out, root_hash[last], 32);
        return 0;
    }
    uint8_t payload[NODE_MAX];
    size_t w = 0;
    for (int b = 0; b < 16; b++) {
        if (child_lens[b]) {
            memcpy(payload + w, child_refs[b], child_lens[b]);
            w += child_lens[b];
        } else {
            payload[w++] = 0x80;
        }
    }
    payload[w++] = 0x80;
    uint8_t rlp[NODE_MAX];
    size_t h = rlp_list_hdr(w, rlp);
    memcpy(rlp + h, payload, w);
    okc_keccak256(rlp, h + w, out);
    return 0;
}

/* ---- multi-threaded state root (OpenMP) ------------------------------
 * The "one thread per core" CPU baseline BASELINE.md's plan names: the
 * same algorithm, storage-trie roots parallel over accounts and the
 * account trie split by top nibble (the okc_subtree_roots decomposition)
 * then finished with okc_finish_top. Used ONLY by bench.py's cpu_baseline
 * leg — test infrastructure, never the product path. */
#ifdef _OPENMP
#include <omp.h>
#endif

int okc_state_root_par(const sre_account_entry *acct, uint64_t na,
                       const sre_storage_entry *st, uint64_t ns,
                       int nthreads, uint8_t out[32])
{
    if (na == 0) {
        if (ns != 0)
            return 3;
        memcpy(out, EMPTY_ROOT, 32);
        return 0;
    }
    for (uint64_t i = 1; i < na; i++)
        if (memcmp(acct[i - 1].key, acct[i].key, 32) >= 0)
            return 1;
    for (uint64_t i = 1; i < ns; i++) {
        int c = memcmp(st[i - 1].acct_key, st[i].acct_key, 32);
        if (c > 0 ||
            (c == 0 && memcmp(st[i - 1].slot_key, st[i].slot_key, 32) >= 0))
            return 2;
    }
    for (uint64_t i = 0; i < ns; i++)
        if (is_zero32(st[i].value))
            return 4;
    uint8_t *roots = (uint8_t *)malloc(na * 32);
    uint64_t *seg_lo = (uint64_t *)malloc(na * sizeof(uint64_t));
    uint64_t *seg_hi = (uint64_t *)malloc(na * sizeof(uint64_t));
    if (!roots || !seg_lo || !seg_hi) {
        free(roots);
        free(seg_lo);
        free(seg_hi);
        return 5;
    }
    /* serial lockstep pass: per-account storage segment bounds */
    uint64_t i = 0;
    int rc = 0;
    for (uint64_t j = 0; j < na; j++) {
        if (i < ns && memcmp(st[i].acct_key, acct[j].key, 32) < 0) {
            rc = 3;
            break;
        }
        seg_lo[j] = i;
        while (i < ns && memcmp(st[i].acct_key, acct[j].key, 32) == 0)
            i++;
        seg_hi[j] = i;
    }
    if (rc == 0 && i != ns)
        rc = 3;
    if (rc) {
        free(roots);
        free(seg_lo);
        free(seg_hi);
        return rc;
    }
#ifdef _OPENMP
    if (nthreads > 0)
        omp_set_num_threads(nthreads);
#else
    (void)nthreads;
#endif
#pragma omp parallel for schedule(dynamic, 64)
    for (uint64_t j = 0; j < na; j++) {
        if (seg_lo[j] == seg_hi[j]) {
            memcpy(roots + 32 * j, EMPTY_ROOT, 32);
        } else {
            trie_src src;
            src.upd = NULL;
            src.key_stride = sizeof(sre_storage_entry);
            src.value = storage_value;
            src.keys = st[seg_lo[j]].slot_key;
            src.vctx = st + seg_lo[j];
            uint8_t rlp[NODE_MAX];
            ref_t ref;
            size_t len = build(&src, 0, seg_hi[j] - seg_lo[j], 0, rlp, &ref,
                               NULL);
            okc_keccak256(rlp, len, roots + 32 * j);
        }
    }
    /* account trie: 16-way top-nibble split, then the shard finisher */
    uint8_t child_refs[16][33], child_lens[16], root_hash[16][32];
    uint64_t counts[16], starts[17];
    memset(child_lens, 0, sizeof(child_lens));
    memset(counts, 0, sizeof(counts));
    {
        uint64_t p = 0;
        for (int b = 0; b < 16; b++) {
            starts[b] = p;
            while (p < na && nib(acct[p].key, 0) == (uint8_t)b)
                p++;
        }
        starts[16] = na;
    }
#pragma omp parallel for schedule(dynamic, 1)
    for (int b = 0; b < 16; b++) {
        uint64_t lo = starts[b], hi = starts[b + 1];
        if (lo == hi)
            continue;
        acct_vctx vc = { acct + lo, roots + 32 * lo };
        trie_src src;
        src.upd = NULL;
        src.keys = acct[lo].key;
        src.key_stride = sizeof(sre_account_entry);
        src.value = account_value;
        src.vctx = &vc;
        uint8_t rlp[NODE_MAX];
        ref_t ref;
        build(&src, 0, hi - lo, 1, rlp, &ref, NULL);
        memcpy(child_refs[b], ref.b, ref.len);
        child_lens[b] = ref.len;
        size_t len = build(&src, 0, hi - lo, 0, rlp, &ref, NULL);
        okc_keccak256(rlp, len, root_hash[b]);
        counts[b] = hi - lo;
    }
    free(seg_lo);
    free(seg_hi);
    free(roots);
    return okc_finish_top(child_refs, child_lens, root_hash, counts, out);
}
