#!/usr/bin/env python3
"""Generate committed golden fixtures from the reference's consensus-pinned data.

Run IN THE BUILD CONTAINER (where /root/reference is mounted); the produced
fixtures are committed so that tests never read /root/reference at run time
(the GPU box does not mount it).

Sources (all public consensus data, no reference CODE is copied):
  - /root/reference/crates/chainspec/res/genesis/{mainnet,sepolia,holesky,dev}.json
    genesis allocs + their consensus-pinned stateRoot fields
  - the fixed 4-slot storage vector and 6-account trie of
    /root/reference/crates/trie/db/tests/trie.rs:384-519 (inputs restated here;
    the expected roots are re-derived by oracle/pyref.py, and for the
    account+storage trie additionally pinned by the hard-coded constant
    0x72861041bc90cd2f93777956f058a545412b56de79af5eb6b8075fe2eabbe015 at
    trie.rs:489).

Output: tests/golden/genesis_<name>.json.gz with
  {"state_root": hex, "accounts": [[address_hex, nonce, balance_hex,
    code_hex_or_null, {slot_hex: value_hex}], ...]}
and tests/golden/fixed_vectors.json.
"""
import gzip
import json
import os
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, os.path.join(HERE, "..", ".."))

from oracle import pyref  # noqa: E402

REF = "/root/reference/crates/chainspec/res/genesis"


def parse_int(s):
    if isinstance(s, int):
        return s
    s = s.strip()
    if s.startswith("0x") or s.startswith("0X"):
        return int(s, 16)
    return int(s)


def main():
    # dev.json is excluded: its stateRoot field equals sepolia's while its
    # alloc has 5 extra accounts — the recorded field is stale in the
    # reference snapshot and does not pin its own alloc.
    for name in ["mainnet", "sepolia", "holesky"]:
        src = os.path.join(REF, f"{name}.json")
        d = json.load(open(src))
        out = {"state_root": d["stateRoot"], "accounts": []}
        for addr, acct in d["alloc"].items():
            a = addr[2:] if addr.startswith("0x") else addr
            nonce = parse_int(acct.get("nonce", 0))
            balance = parse_int(acct.get("balance", 0))
            code = acct.get("code")
            if code is not None:
                code = code[2:] if code.startswith("0x") else code
            storage = {}
            for slot, val in acct.get("storage", {}).items():
                s = slot[2:] if slot.startswith("0x") else slot
                storage[s.lower().rjust(64, "0")] = val
            out["accounts"].append([a.lower(), nonce, hex(balance), code, storage])
        # verify with pyref before writing
        accounts = {}
        for a, nonce, bal, code, storage in out["accounts"]:
            haddr = pyref.keccak256(bytes.fromhex(a))
            code_hash = pyref.keccak256(bytes.fromhex(code)) if code else pyref.KECCAK_EMPTY
            slots = {pyref.keccak256(bytes.fromhex(s)): parse_int(v) for s, v in storage.items()}
            accounts[haddr] = (nonce, parse_int(bal), code_hash, slots)
        got = pyref.state_root(accounts)
        want = out["state_root"]
        status = "OK" if "0x" + got.hex() == want else "MISMATCH"
        print(f"{name}: pyref={got.hex()} expected={want} {status}")
        if status != "OK":
            raise SystemExit(f"pyref does not reproduce {name} genesis root")
        with gzip.open(os.path.join(HERE, f"genesis_{name}.json.gz"), "wt") as f:
            json.dump(out, f)

    # Fixed vectors restated from /root/reference/crates/trie/db/tests/trie.rs:384-519
    storage4 = {
        "1200000000000000000000000000000000000000000000000000000000000000": 0x42,
        "1400000000000000000000000000000000000000000000000000000000000000": 0x01,
        "3000000000000000000000000000000000000000000000000000000000E00000": 0x127A89,
        "3000000000000000000000000000000000000000000000000000000000E00001": 0x05,
    }
    # storage_root_prehashed: slots are used as ALREADY-HASHED keys
    sr = pyref.trie_root({bytes.fromhex(k): pyref.rlp_int(v) for k, v in storage4.items()})
    ether = 10 ** 18
    code_hash3 = "5be74cad16203c4905c068b012a2e9fb6d19d036c410f16fd177f337541440dd"
    accts = [
        # (prehashed key, nonce, balance, code_hash, storage_root)
        ("b000000000000000000000000000000000000000000000000000000000000000", 0, 3 * ether, None, None),
        (pyref.keccak256(bytes.fromhex("7db3e81b72d2695e19764583f6d219dbee0f35ca")).hex(), 0, ether, None, None),
        (pyref.keccak256(bytes.fromhex("16b07afd1c635f77172e842a000ead9a2a222459")).hex(), 0, 2 * ether, code_hash3, sr.hex()),
        ("b1a0000000000000000000000000000000000000000000000000000000000000", 0, 4 * ether, None, None),
        ("b310000000000000000000000000000000000000000000000000000000000000", 0, 8 * ether, None, None),
        ("b340000000000000000000000000000000000000000000000000000000000000", 0, 1 * ether, None, None),
    ]
    items = {}
    for hk, nonce, bal, ch, srh in accts:
        items[bytes.fromhex(hk)] = pyref.account_value(
            nonce, bal,
            bytes.fromhex(srh) if srh else pyref.EMPTY_ROOT_HASH,
            bytes.fromhex(ch) if ch else pyref.KECCAK_EMPTY)
    root = pyref.trie_root(items)
    expected = "72861041bc90cd2f93777956f058a545412b56de79af5eb6b8075fe2eabbe015"
    print(f"account_and_storage_trie: pyref={root.hex()} expected={expected} "
          f"{'OK' if root.hex() == expected else 'MISMATCH'}")
    if root.hex() != expected:
        raise SystemExit("pyref does not reproduce trie.rs:489 pinned root")
    fixed = {
        "storage_root_regression": {
            "prehashed_slots": {k.lower(): hex(v) for k, v in storage4.items()},
            "storage_root": sr.hex(),
        },
        "account_and_storage_trie": {
            "accounts": [[hk, n, hex(b), ch, srh] for hk, n, b, ch, srh in accts],
            "state_root": expected,
        },
    }
    with open(os.path.join(HERE, "fixed_vectors.json"), "w") as f:
        json.dump(fixed, f, indent=1)
    print("fixtures written")


if __name__ == "__main__":
    main()
