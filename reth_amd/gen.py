"""Deterministic synthetic-state generator (seed 0x5EED).

Spec (identical for the numpy/CPU and torch backends — equality is pinned by
tests/test_gen_cpu.py on CPU and by the GPU parity tests):

  account counter c in [0, n_accounts):
    hashed key   = keccak256(LE64(c))
    nonce        = c & 0xFFFF
    balance      = 16-byte big-endian value: hi = splitmix(c ^ S_BAL_HI) | 1,
                   lo = splitmix(c ^ S_BAL_LO)   (nonzero by construction)
    code_hash    = KECCAK_EMPTY when splitmix(c ^ S_CODE) % 10 < 9,
                   else 32 bytes from LE64(splitmix(c ^ S_CH[k])) k=0..3
  slot j in [0, slots) of account c:
    hashed slot key = keccak256(LE64(c) || LE64(j))
    value: minimal-BE length L = 1 + (splitmix(x) & 31) with
           x = (c << 32 | j) ^ S_VAL; top byte 1 + splitmix(x ^ S_VB) % 255;
           lower bytes from LE64 chunks of splitmix(x ^ S_VK[k])
           (exercises every RLP length branch: 1-byte <0x80 .. 32 B)

splitmix(x) = splitmix64 seeded by folding SEED into the gamma constant.
Entries sorted by hashed key (accounts) / (account position, slot key).
"""
import numpy as np

from reth_amd.engine import ACCOUNT_DTYPE, STORAGE_DTYPE

SEED = 0x5EED
KECCAK_EMPTY = bytes.fromhex(
    "c5d2460186f7233c927e7db2dcc703c0e500b653ca82273b7bfad8045d85a470")

M64 = (1 << 64) - 1
GAMMA = (0x9E3779B97F4A7C15 ^ (SEED * 0x2545F4914F6CDD1D)) & M64
S_BAL_HI = 0xB1AC5EED00000001
S_BAL_LO = 0xB1AC5EED00000002
S_CODE = 0xB1AC5EED00000003
S_CH = [0xB1AC5EED0000C000 + k for k in range(4)]
S_VAL = 0xB1AC5EED00000004
S_VB = 0xB1AC5EED00000005
S_VK = [0xB1AC5EED0000D000 + k for k in range(4)]


# ---------------------------------------------------------------------------
# numpy backend (CPU: parity tests, small configs, cpu_baseline sampling)
# ---------------------------------------------------------------------------

def _sm64_np(x):
    z = (x + np.uint64(GAMMA)) & np.uint64(M64)
    z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
    z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
    return z ^ (z >> np.uint64(31))


def _np_le64(x):
    return x.astype("<u8").view(np.uint8).reshape(-1, 8)


def _np_be64(x):
    return x.astype(">u8").view(np.uint8).reshape(-1, 8)


def _np_sort_keys(keys):
    """(n,32) u8 big-endian keys -> stable ascending permutation."""
    return np.lexsort(tuple(
        keys[:, 8 * k:8 * (k + 1)].copy().view(">u8").ravel()
        for k in reversed(range(4))))


def gen_state_numpy(n_accounts, slots_per_account, keccak_batch,
                    nibble_filter=None):
    """keccak_batch: (msgs (n,L) u8) -> (n,32) digests (e.g. oracle.bind's).
    nibble_filter: optional callable(top_nibbles int64 array) -> bool mask
    over the GLOBAL account set (multi-rank sharding)."""
    old = np.seterr(over="ignore")
    try:
        c = np.arange(n_accounts, dtype=np.uint64)
        keys = keccak_batch(_np_le64(c))
        if nibble_filter is not None:
            mask = nibble_filter((keys[:, 0] >> 4).astype(np.int64))
            c, keys = c[mask], keys[mask]
        na = len(c)
        order = _np_sort_keys(keys)
        cs = c[order]
        acct = np.zeros(na, dtype=ACCOUNT_DTYPE)
        acct["key"] = keys[order]
        acct["nonce"] = cs & np.uint64(0xFFFF)
        hi = _sm64_np(cs ^ np.uint64(S_BAL_HI)) | np.uint64(1)
        lo = _sm64_np(cs ^ np.uint64(S_BAL_LO))
        bal = np.zeros((na, 32), dtype=np.uint8)
        bal[:, 16:24] = _np_be64(hi)
        bal[:, 24:32] = _np_be64(lo)
        acct["balance"] = bal
        ch = np.tile(np.frombuffer(KECCAK_EMPTY, np.uint8), (na, 1)).copy()
        rare = (_sm64_np(cs ^ np.uint64(S_CODE)) % np.uint64(10)) >= np.uint64(9)
        for k in range(4):
            v = _np_le64(_sm64_np(cs ^ np.uint64(S_CH[k])))
            ch[rare, 8 * k:8 * (k + 1)] = v[rare]
        acct["code_hash"] = ch

        ns = na * slots_per_account
        st = np.zeros(ns, dtype=STORAGE_DTYPE)
        if ns:
            j = np.arange(slots_per_account, dtype=np.uint64)
            cc = np.repeat(cs, slots_per_account)
            jj = np.tile(j, na)
            msg = np.zeros((ns, 16), dtype=np.uint8)
            msg[:, :8] = _np_le64(cc)
            msg[:, 8:] = _np_le64(jj)
            skeys = keccak_batch(msg)
            # stable sort: slot key minor, account position major
            grp = np.repeat(np.arange(na, dtype=np.uint64), slots_per_account)
            sord = np.lexsort(tuple(
                skeys[:, 8 * k:8 * (k + 1)].copy().view(">u8").ravel()
                for k in reversed(range(4))) + (grp,))
            x = ((cc << np.uint64(32)) | jj) ^ np.uint64(S_VAL)
            L = (np.uint64(1) + (_sm64_np(x) & np.uint64(31))).astype(np.int64)
            top = (np.uint64(1) + _sm64_np(x ^ np.uint64(S_VB)) % np.uint64(255)
                   ).astype(np.uint8)
            body = np.zeros((ns, 32), dtype=np.uint8)
            for k in range(4):
                body[:, 8 * k:8 * (k + 1)] = _np_le64(_sm64_np(x ^ np.uint64(S_VK[k])))
            colim = np.arange(32)[None, :]
            vals = np.where(colim >= (32 - L)[:, None], body, 0).astype(np.uint8)
            vals[np.arange(ns), 32 - L] = top
            st["acct_key"] = np.repeat(acct["key"], slots_per_account, axis=0)[sord]
            st["slot_key"] = skeys[sord]
            st["value"] = vals[sord]
        return acct, st
    finally:
        np.seterr(**old)


# ---------------------------------------------------------------------------
# torch backend (GPU bench-scale generation; also runs on CPU for tests)
# ---------------------------------------------------------------------------

def _i64(v):
    """Python int (unsigned 64-bit) -> int64 two's-complement literal."""
    v &= M64
    return v - (1 << 64) if v >= (1 << 63) else v


def _lsr(x, k):
    """logical right shift on int64 tensors"""
    return (x >> k) & _i64((1 << (64 - k)) - 1) if k else x


def _umod(x, m):
    """unsigned x mod m (m small positive) on int64 tensors"""
    hi = _lsr(x, 32)
    lo = x & 0xFFFFFFFF
    return ((hi % m) * ((1 << 32) % m) + lo % m) % m


def _sm64_t(x):
    z = x + _i64(GAMMA)
    z = (z ^ _lsr(z, 30)) * _i64(0xBF58476D1CE4E5B9)
    z = (z ^ _lsr(z, 27)) * _i64(0x94D049BB133111EB)
    return z ^ _lsr(z, 31)


def _bytes_le(x):
    import torch
    out = torch.empty((x.shape[0], 8), dtype=torch.uint8, device=x.device)
    for k in range(8):
        out[:, k] = (_lsr(x, 8 * k) & 0xFF).to(torch.uint8)
    return out


def _bytes_be(x):
    import torch
    out = torch.empty((x.shape[0], 8), dtype=torch.uint8, device=x.device)
    for k in range(8):
        out[:, 7 - k] = (_lsr(x, 8 * k) & 0xFF).to(torch.uint8)
    return out


def _argsort_keys_t(keys_u8, group=None):
    """Stable ascending sort of (n,32) big-endian byte keys; optional primary
    group key (int64). Returns permutation (int64)."""
    import torch
    n = keys_u8.shape[0]
    order = torch.arange(n, dtype=torch.int64, device=keys_u8.device)
    cols = keys_u8.view(n, 4, 8)
    for k in reversed(range(4)):
        col = cols[:, k, :].to(torch.int64)
        v = torch.zeros(n, dtype=torch.int64, device=keys_u8.device)
        for b in range(8):
            v = (v << 8) | col[:, b]
        v = v ^ _i64(1 << 63)  # unsigned order under int64 compare
        perm = torch.argsort(v[order], stable=True)
        order = order[perm]
    if group is not None:
        perm = torch.argsort(group[order], stable=True)
        order = order[perm]
    return order


def gen_state_torch(n_accounts, slots_per_account, keccak_batch_device,
                    device="cuda", nibble_filter=None):
    """Returns (acct (na,104) u8, st (ns,96) u8) tensors laid out as
    sre_account_entry / sre_storage_entry, sorted per the input contract.
    keccak_batch_device: (in_u8 (n,L), L, out (n,32)) -> out."""
    import torch
    c_all = torch.arange(n_accounts, dtype=torch.int64, device=device)
    keys_all = torch.empty((n_accounts, 32), dtype=torch.uint8, device=device)
    keccak_batch_device(_bytes_le(c_all), 8, keys_all)
    if nibble_filter is not None:
        mask = nibble_filter((keys_all[:, 0] >> 4).to(torch.int64))
        c = c_all[mask]
        keys = keys_all[mask].contiguous()
    else:
        c, keys = c_all, keys_all
    na = c.shape[0]
    order = _argsort_keys_t(keys)
    cs = c[order]
    keys = keys[order].contiguous()

    acct = torch.zeros((na, 104), dtype=torch.uint8, device=device)
    acct[:, 0:32] = keys
    acct[:, 32:40] = _bytes_le(cs & 0xFFFF)
    hi = _sm64_t(cs ^ _i64(S_BAL_HI)) | 1
    lo = _sm64_t(cs ^ _i64(S_BAL_LO))
    acct[:, 56:64] = _bytes_be(hi)
    acct[:, 64:72] = _bytes_be(lo)
    ke = torch.tensor(list(KECCAK_EMPTY), dtype=torch.uint8, device=device)
    ch = ke.repeat(na, 1)
    rare = _umod(_sm64_t(cs ^ _i64(S_CODE)), 10) >= 9
    for k in range(4):
        v = _bytes_le(_sm64_t(cs ^ _i64(S_CH[k])))
        ch[:, 8 * k:8 * (k + 1)] = torch.where(rare[:, None], v,
                                               ch[:, 8 * k:8 * (k + 1)])
    acct[:, 72:104] = ch

    ns = na * slots_per_account
    st = torch.zeros((ns, 96), dtype=torch.uint8, device=device)
    if ns:
        j = torch.arange(slots_per_account, dtype=torch.int64, device=device)
        cc = cs.repeat_interleave(slots_per_account)
        jj = j.repeat(na)
        msg = torch.cat([_bytes_le(cc), _bytes_le(jj)], dim=1).contiguous()
        skeys = torch.empty((ns, 32), dtype=torch.uint8, device=device)
        keccak_batch_device(msg, 16, skeys)
        del msg
        grp = torch.arange(na, dtype=torch.int64, device=device) \
                   .repeat_interleave(slots_per_account)
        sord = _argsort_keys_t(skeys, group=grp)
        del grp
        x = ((cc << 32) | jj) ^ _i64(S_VAL)
        L = (1 + (_sm64_t(x) & 31)).to(torch.int64)
        top = (1 + _umod(_sm64_t(x ^ _i64(S_VB)), 255)).to(torch.uint8)
        body = torch.cat([_bytes_le(_sm64_t(x ^ _i64(S_VK[k]))) for k in range(4)],
                         dim=1)
        del cc, jj, x
        colim = torch.arange(32, device=device)[None, :]
        vals = torch.where(colim >= (32 - L)[:, None], body,
                           torch.zeros_like(body))
        del body
        vals[torch.arange(ns, device=device), 32 - L] = top
        del top, L
        st[:, 0:32] = acct[:, 0:32].repeat_interleave(slots_per_account, dim=0)[sord]
        st[:, 32:64] = skeys[sord]
        del skeys
        st[:, 64:96] = vals[sord]
        del vals, sord
    return acct, st


def np_state_to_tensors(acct, st, device="cpu"):
    """numpy structured arrays -> raw (na,104)/(ns,96) uint8 tensors."""
    import torch
    a = torch.from_numpy(acct.view(np.uint8).reshape(len(acct), 104).copy())
    s = torch.from_numpy(st.view(np.uint8).reshape(len(st), 96).copy())
    return a.to(device), s.to(device)


def tensors_to_np_state(acct_u8, st_u8):
    """raw uint8 tensors (any device) -> numpy structured arrays."""
    a = acct_u8.cpu().numpy().reshape(-1).view(ACCOUNT_DTYPE)
    s = st_u8.cpu().numpy().reshape(-1).view(STORAGE_DTYPE)
    return a, s
