"""Pure-Python Merkle Patricia Trie reference — TEST INFRASTRUCTURE ONLY.

This module is an independent restatement of the Ethereum secure-MPT state
commitment that reth's state-root path computes:

  - `StateRoot::calculate` semantics: /root/reference/crates/trie/trie/src/trie.rs:171-374
  - leaf node = RLP([HP(short_key, leaf), RLP_value]), ref = RLP if len<32 else
    0xa0||keccak(RLP): /root/reference/crates/trie/trie/src/proof_v2/node.rs:40-66
  - storage leaf value = encode_fixed_size(U256) (minimal big-endian RLP):
    /root/reference/crates/trie/trie/src/trie.rs:819-825, proof_v2/value.rs:55-66
  - account leaf value = RLP([nonce, balance, storage_root, code_hash]) (<=110 B):
    /root/reference/crates/trie/trie/src/trie.rs:472-476,
    /root/reference/crates/trie/common/src/root.rs:9-31, proof_v2/value.rs:125-139
  - hex-prefix (compact) encoding and branch/extension shapes are spec-only in
    the reference (they live in the external alloy-trie/nybbles crates; shapes
    confirmed by /root/reference/crates/trie/trie/src/node_iter.rs:420-449):
    restated here from the Yellow Paper.

It is deliberately written in the most naive recursive style (dict in,
recursion over nibble partitions) so that it shares no structure with either
the C oracle (sorted-stream recursion, oracle/mpt_oracle.c) or the HIP engine
(bottom-up LCP levels, reth_amd/csrc). Only `tests/`, golden-fixture
generation, and oracle cross-checks may import this module. It must never be
on a product path or inside a timed region.

Pinned against consensus data: the genesis stateRoot vectors in
/root/reference/crates/chainspec/res/genesis/{mainnet,sepolia,holesky,dev}.json
(fixtures committed under tests/golden/) and the hard-coded root of
/root/reference/crates/trie/db/tests/trie.rs:420-519.
"""

# --------------------------------------------------------------------------
# Keccak-256 (original Keccak padding 0x01, NOT FIPS-202 SHA3's 0x06).
# Independent of the C oracle's implementation.
# --------------------------------------------------------------------------

_ROT = [
    [0, 36, 3, 41, 18],
    [1, 44, 10, 45, 2],
    [62, 6, 43, 15, 61],
    [28, 55, 25, 21, 56],
    [27, 20, 39, 8, 14],
]

_RC = [
    0x0000000000000001, 0x0000000000008082, 0x800000000000808A, 0x8000000080008000,
    0x000000000000808B, 0x0000000080000001, 0x8000000080008081, 0x8000000000008009,
    0x000000000000008A, 0x0000000000000088, 0x0000000080008009, 0x000000008000000A,
    0x000000008000808B, 0x800000000000008B, 0x8000000000008089, 0x8000000000008003,
    0x8000000000008002, 0x8000000000000080, 0x000000000000800A, 0x800000008000000A,
    0x8000000080008081, 0x8000000000008080, 0x0000000080000001, 0x8000000080008008,
]

_M64 = (1 << 64) - 1


def _rotl(x, n):
    return ((x << n) | (x >> (64 - n))) & _M64


def _keccak_f(a):
    for rc in _RC:
        # theta
        c = [a[x][0] ^ a[x][1] ^ a[x][2] ^ a[x][3] ^ a[x][4] for x in range(5)]
        d = [c[(x - 1) % 5] ^ _rotl(c[(x + 1) % 5], 1) for x in range(5)]
        for x in range(5):
            for y in range(5):
                a[x][y] ^= d[x]
        # rho + pi
        b = [[0] * 5 for _ in range(5)]
        for x in range(5):
            for y in range(5):
                b[y][(2 * x + 3 * y) % 5] = _rotl(a[x][y], _ROT[x][y])
        # chi
        for x in range(5):
            for y in range(5):
                a[x][y] = b[x][y] ^ ((~b[(x + 1) % 5][y]) & b[(x + 2) % 5][y])
        # iota
        a[0][0] ^= rc


def keccak256(data: bytes) -> bytes:
    rate = 136
    a = [[0] * 5 for _ in range(5)]
    # pad10*1 with Keccak domain bit 0x01
    padded = bytearray(data)
    padded.append(0x01)
    while len(padded) % rate != 0:
        padded.append(0x00)
    padded[-1] |= 0x80
    for off in range(0, len(padded), rate):
        block = padded[off:off + rate]
        for i in range(rate // 8):
            lane = int.from_bytes(block[8 * i:8 * i + 8], "little")
            x, y = i % 5, i // 5
            a[x][y] ^= lane
        _keccak_f(a)
    out = bytearray()
    for i in range(4):  # 32 bytes = 4 lanes
        x, y = i % 5, i // 5
        out += a[x][y].to_bytes(8, "little")
    return bytes(out)


KECCAK_EMPTY = keccak256(b"")
EMPTY_ROOT_HASH = keccak256(b"\x80")  # keccak(rlp(""))


# --------------------------------------------------------------------------
# RLP
# --------------------------------------------------------------------------

def rlp_str(b: bytes) -> bytes:
    if len(b) == 1 and b[0] < 0x80:
        return b
    if len(b) < 56:
        return bytes([0x80 + len(b)]) + b
    lb = len(b).to_bytes((len(b).bit_length() + 7) // 8, "big")
    return bytes([0xB7 + len(lb)]) + lb + b


def rlp_list_payload(payload: bytes) -> bytes:
    if len(payload) < 56:
        return bytes([0xC0 + len(payload)]) + payload
    lb = len(payload).to_bytes((len(payload).bit_length() + 7) // 8, "big")
    return bytes([0xF7 + len(lb)]) + lb + payload


def rlp_int(v: int) -> bytes:
    if v == 0:
        return b"\x80"
    return rlp_str(v.to_bytes((v.bit_length() + 7) // 8, "big"))


# --------------------------------------------------------------------------
# MPT
# --------------------------------------------------------------------------

def hp_encode(nibbles, is_leaf: bool) -> bytes:
    odd = len(nibbles) % 2
    first = (0x20 if is_leaf else 0x00) | (0x10 if odd else 0x00)
    out = bytearray()
    if odd:
        out.append(first | nibbles[0])
        rest = nibbles[1:]
    else:
        out.append(first)
        rest = nibbles
    for i in range(0, len(rest), 2):
        out.append((rest[i] << 4) | rest[i + 1])
    return bytes(out)


def nibbles_of(key: bytes):
    out = []
    for b in key:
        out.append(b >> 4)
        out.append(b & 0x0F)
    return tuple(out)


def _node_ref(rlp: bytes) -> bytes:
    """Child reference as embedded in the parent's RLP payload."""
    if len(rlp) < 32:
        return rlp
    return b"\xa0" + keccak256(rlp)


def _build(items, pos):
    """items: sorted list of (nibble_tuple, value_bytes); returns node RLP.

    Caller guarantees len(items) >= 1 and all keys share a prefix of length
    >= pos ... actually exactly: all keys agree on [0, pos).
    """
    if len(items) == 1:
        nib, val = items[0]
        return rlp_list_payload(rlp_str(hp_encode(nib[pos:], True)) + rlp_str(val))
    # common prefix of the range (sorted => lcp(first, last))
    first, last = items[0][0], items[-1][0]
    p = pos
    while p < len(first) and p < len(last) and first[p] == last[p]:
        p += 1
    if p > pos:
        child = _build(items, p)
        payload = rlp_str(hp_encode(first[pos:p], False)) + _node_ref(child)
        return rlp_list_payload(payload)
    # branch at pos
    payload = b""
    i = 0
    for nib in range(16):
        j = i
        while j < len(items) and items[j][0][pos] == nib:
            j += 1
        if j == i:
            payload += b"\x80"
        else:
            child = _build(items[i:j], pos + 1)
            payload += _node_ref(child)
            i = j
    payload += b"\x80"  # 17th (value) slot: always empty in state/storage tries
    return rlp_list_payload(payload)


def trie_root(items: dict) -> bytes:
    """items: {key_bytes: value_bytes} (value = RLP-encoded leaf value)."""
    if not items:
        return EMPTY_ROOT_HASH
    lst = sorted((nibbles_of(k), v) for k, v in items.items())
    return keccak256(_build(lst, 0))


def storage_root(slots: dict) -> bytes:
    """slots: {hashed_slot_32B: int_value}; zero values must be absent."""
    items = {k: rlp_int(v) for k, v in slots.items() if v != 0}
    return trie_root(items)


def account_value(nonce: int, balance: int, storage_root_: bytes, code_hash: bytes) -> bytes:
    payload = rlp_int(nonce) + rlp_int(balance) + rlp_str(storage_root_) + rlp_str(code_hash)
    return rlp_list_payload(payload)


def state_root(accounts: dict) -> bytes:
    """accounts: {hashed_address_32B: (nonce, balance, code_hash_32B, {hashed_slot: int})}."""
    items = {}
    for k, (nonce, balance, code_hash, slots) in accounts.items():
        sr = storage_root(slots)
        items[k] = account_value(nonce, balance, sr, code_hash)
    return trie_root(items)


def _collect_proof(items, pos, target, is_root, nodes):
    """Walk the _build recursion along `target` (nibble tuple), appending
    every node RLP that is hash-referenced (>= 32 B) or the root — the
    standard eth_getProof node-list semantics (the reference's
    ProofRetainer keyed by path prefix, crates/trie/trie/src/proof/mod.rs)."""
    rlp = _build(items, pos)
    if len(items) == 1:
        if is_root or len(rlp) >= 32:
            nodes.append(rlp)
        return
    first, last = items[0][0], items[-1][0]
    p = pos
    while p < len(first) and p < len(last) and first[p] == last[p]:
        p += 1
    if p > pos:
        # extension node, then its branch child
        if is_root or len(rlp) >= 32:
            nodes.append(rlp)
        if target[pos:p] != first[pos:p]:
            return  # extension path diverges: absence proven here
        brlp = _build(items, p)
        if len(brlp) >= 32:
            nodes.append(brlp)
        _descend_branch(items, p, target, nodes)
        return
    if is_root or len(rlp) >= 32:
        nodes.append(rlp)
    _descend_branch(items, pos, target, nodes)


def _descend_branch(items, pos, target, nodes):
    i = 0
    for nib in range(16):
        j = i
        while j < len(items) and items[j][0][pos] == nib:
            j += 1
        if nib == target[pos]:
            if j == i:
                return  # empty child slot: absence proven at this branch
            _collect_proof(items[i:j], pos + 1, target, False, nodes)
            return
        i = j
    raise AssertionError("unreachable")


def account_proof(accounts: dict, hashed_key: bytes):
    """Proof node list (root-first RLPs) for a hashed account key — the
    lookup-path nodes; for an absent key the list ends at the divergence
    (exclusion proof). Empty state => empty list."""
    items_d = {}
    for k, (nonce, balance, code_hash, slots) in accounts.items():
        sr = storage_root(slots)
        items_d[k] = account_value(nonce, balance, sr, code_hash)
    items = sorted((tuple(nibbles_of(k)), v) for k, v in items_d.items())
    target = tuple(nibbles_of(hashed_key))
    nodes = []
    if items:
        _collect_proof(items, 0, target, True, nodes)
    return nodes


def storage_proof(accounts: dict, acct_key: bytes, slot_key: bytes):
    """(storage_root, proof-node list root-first) for a slot —
    StorageProof::storage_multiproof semantics. The slot may be absent:
    _collect_proof then returns the lookup-path nodes ending at the
    proven divergence (an exclusion proof). An absent or storage-less
    account yields (EMPTY_ROOT_HASH, []) = StorageMultiProof::empty(),
    matching engine.storage_proof / include/sre.h."""
    if acct_key not in accounts:
        return keccak256(b"\x80"), []
    slots = accounts[acct_key][3]
    items = sorted((tuple(nibbles_of(k)), rlp_int(v))
                   for k, v in slots.items() if v != 0)
    target = tuple(nibbles_of(slot_key))
    nodes = []
    if items:
        _collect_proof(items, 0, target, True, nodes)
        root = keccak256(_build(items, 0))
    else:
        root = keccak256(b"\x80")  # EMPTY_ROOT_HASH
    return root, nodes
