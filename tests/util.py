"""Shared helpers for tests: golden-fixture loading and array conversion."""
import gzip
import json
import os

import numpy as np

from oracle import bind, pyref

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def load_genesis(name):
    """Returns ({hashed_addr: (nonce, balance, code_hash, {hashed_slot: int})}, root_hex)."""
    d = json.load(gzip.open(os.path.join(GOLDEN, f"genesis_{name}.json.gz"), "rt"))
    accounts = {}
    for a, nonce, bal, code, storage in d["accounts"]:
        haddr = bind.keccak256(bytes.fromhex(a))
        ch = bind.keccak256(bytes.fromhex(code)) if code else pyref.KECCAK_EMPTY
        slots = {bind.keccak256(bytes.fromhex(s)): int(v, 16) for s, v in storage.items()}
        accounts[haddr] = (nonce, int(bal, 16), ch, slots)
    return accounts, d["state_root"]


def to_arrays(accounts):
    """accounts dict -> (sre_account_entry array, sre_storage_entry array), sorted."""
    acct = np.zeros(len(accounts), dtype=bind.ACCOUNT_DTYPE)
    st_rows = []
    for i, (k, (nonce, bal, ch, slots)) in enumerate(sorted(accounts.items())):
        acct[i]["key"] = np.frombuffer(k, np.uint8)
        acct[i]["nonce"] = nonce
        acct[i]["balance"] = np.frombuffer(bal.to_bytes(32, "big"), np.uint8)
        acct[i]["code_hash"] = np.frombuffer(ch, np.uint8)
        for sk, v in sorted(slots.items()):
            st_rows.append((k, sk, v))
    st = np.zeros(len(st_rows), dtype=bind.STORAGE_DTYPE)
    for i, (a, s, v) in enumerate(st_rows):
        st[i]["acct_key"] = np.frombuffer(a, np.uint8)
        st[i]["slot_key"] = np.frombuffer(s, np.uint8)
        st[i]["value"] = np.frombuffer(v.to_bytes(32, "big"), np.uint8)
    return acct, st


def random_accounts(trial, n, max_slots=7, single_nibble=False, seed=0):
    """Deterministic random account set for cross-checks."""
    import random
    rng = random.Random((seed << 32) ^ trial)
    accounts = {}
    for i in range(n):
        k = bind.keccak256(i.to_bytes(8, "little") + trial.to_bytes(4, "little"))
        if single_nibble:
            k = bytes([0x30]) + k[1:]
        nslots = rng.choice([0, 0, 1, 2, max_slots])
        slots = {}
        for j in range(nslots):
            sk = bind.keccak256(b"slot" + i.to_bytes(4, "little") + j.to_bytes(4, "little"))
            v = rng.getrandbits(rng.choice([7, 8, 64, 255]))
            if v:
                slots[sk] = v
        accounts[k] = (rng.getrandbits(16), rng.getrandbits(100),
                       pyref.KECCAK_EMPTY if rng.random() < 0.9
                       else bind.keccak256(b"code" + bytes([i % 251])),
                       slots)
    return accounts
