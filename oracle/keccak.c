/* keccak.c — portable scalar Keccak-256 for the CPU oracle.
 *
 * TEST INFRASTRUCTURE + REPORTED CPU BASELINE ONLY: this file restates the
 * keccak256 the reference calls through alloy-primitives -> keccak-asm 0.1.6
 * (CRYPTOGAMS SHA-3 assembly; call sites
 * /root/reference/crates/stages/stages/src/stages/hashing_account.rs:201,
 * hashing_storage.rs:133-137, crates/trie/common/src/hashed_state.rs:49-70,
 * and every node hash inside alloy-trie's HashBuilder). Original Keccak
 * padding (domain byte 0x01), rate 136 — NOT FIPS-202 SHA3-256 (0x06).
 *
 * Only oracle/, tests/, __graft_entry__.smoke() and bench.py's cpu_baseline
 * leg may link or call this code. The product path (libsre) has its own HIP
 * device implementation and must never route here.
 */
#include <stddef.h>
#include <stdint.h>
#include <string.h>

#define ROTL64(x, n) (((x) << (n)) | ((x) >> (64 - (n))))

static const uint64_t RC[24] = {
    0x0000000000000001ULL, 0x0000000000008082ULL, 0x800000000000808aULL,
    0x8000000080008000ULL, 0x000000000000808bULL, 0x0000000080000001ULL,
    0x8000000080008081ULL, 0x8000000000008009ULL, 0x000000000000008aULL,
    0x0000000000000088ULL, 0x0000000080008009ULL, 0x000000008000000aULL,
    0x000000008000808bULL, 0x800000000000008bULL, 0x8000000000008089ULL,
    0x8000000000008003ULL, 0x8000000000008002ULL, 0x8000000000000080ULL,
    0x000000000000800aULL, 0x800000008000000aULL, 0x8000000080008081ULL,
    0x8000000000008080ULL, 0x0000000080000001ULL, 0x8000000080008008ULL,
};

/* One Keccak-f[1600] permutation, fully unrolled x/y loops per round. */
void okc_keccak_f1600(uint64_t s[25])
{
    uint64_t b[25], c[5], d[5];
    for (int r = 0; r < 24; r++) {
        /* theta */
        for (int x = 0; x < 5; x++)
            c[x] = s[x] ^ s[x + 5] ^ s[x + 10] ^ s[x + 15] ^ s[x + 20];
        for (int x = 0; x < 5; x++)
            d[x] = c[(x + 4) % 5] ^ ROTL64(c[(x + 1) % 5], 1);
        for (int i = 0; i < 25; i++)
            s[i] ^= d[i % 5];
        /* rho + pi  (index = x + 5y; B[y][(2x+3y)%5] = rot(A[x][y])) */
        static const uint8_t rot[25] = {
            0, 1, 62, 28, 27, 36, 44, 6, 55, 20, 3, 10, 43, 25, 39,
            41, 45, 15, 21, 8, 18, 2, 61, 56, 14,
        };
        for (int x = 0; x < 5; x++)
            for (int y = 0; y < 5; y++) {
                int src = x + 5 * y;
                int dst = y + 5 * ((2 * x + 3 * y) % 5);
                b[dst] = rot[src] ? ROTL64(s[src], rot[src]) : s[src];
            }
        /* chi */
        for (int y = 0; y < 5; y++)
            for (int x = 0; x < 5; x++)
                s[x + 5 * y] = b[x + 5 * y] ^ ((~b[(x + 1) % 5 + 5 * y]) & b[(x + 2) % 5 + 5 * y]);
        /* iota */
        s[0] ^= RC[r];
    }
}

void okc_keccak256(const uint8_t *in, size_t len, uint8_t out[32])
{
    uint64_t s[25];
    memset(s, 0, sizeof(s));
    const size_t rate = 136;
    /* absorb full blocks */
    while (len >= rate) {
        for (int i = 0; i < 17; i++) {
            uint64_t lane;
            memcpy(&lane, in + 8 * i, 8); /* little-endian host assumed (x86/amdgcn) */
            s[i] ^= lane;
        }
        okc_keccak_f1600(s);
        in += rate;
        len -= rate;
    }
    /* final block with pad10*1, domain 0x01 */
    uint8_t block[136];
    memset(block, 0, sizeof(block));
    memcpy(block, in, len);
    block[len] = 0x01;
    block[135] |= 0x80;
    for (int i = 0; i < 17; i++) {
        uint64_t lane;
        memcpy(&lane, block + 8 * i, 8);
        s[i] ^= lane;
    }
    okc_keccak_f1600(s);
    memcpy(out, s, 32);
}

/* Batch helper: n messages of `len` bytes each at `stride` apart. */
void okc_keccak256_batch(const uint8_t *in, size_t stride, size_t len, size_t n,
                         uint8_t *out)
{
    for (size_t i = 0; i < n; i++)
        okc_keccak256(in + i * stride, len, out + 32 * i);
}
