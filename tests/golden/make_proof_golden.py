#!/usr/bin/env python3
"""Generate committed golden proof vectors (tests/golden/proof_vectors.json)
from the pure-python oracle walker: small deterministic states with
account + storage proofs for present and absent keys. Regenerate with:
    python tests/golden/make_proof_golden.py
"""
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from oracle import bind, pyref  # noqa: E402


def main():
    ke = bind.keccak256(b"")
    accounts = {}
    for i in range(12):
        k = bind.keccak256(b"acct" + bytes([i]))
        slots = {bind.keccak256(b"slot" + bytes([i, q])): (i + 1) * 100 + q
                 for q in range(i % 4)}
        accounts[k] = (i, 10**15 + i, ke if i % 3 else bind.keccak256(b"c"),
                       slots)
    root = pyref.state_root(accounts)
    vectors = {"state_root": root.hex(), "accounts": [], "storage": []}
    keys = sorted(accounts)
    targets = keys[:3] + [bind.keccak256(b"absent" + bytes([j]))
                          for j in range(2)]
    for k in targets:
        nodes = pyref.account_proof(accounts, k)
        vectors["accounts"].append({
            "key": k.hex(),
            "present": k in accounts,
            "nodes": [n.hex() for n in nodes],
        })
    ak = keys[7]  # has slots (7 % 4 == 3)
    sks = sorted(accounts[ak][3])[:2] + [bind.keccak256(b"noslot")]
    for sk in sks:
        r, nodes = pyref.storage_proof(accounts, ak, sk)
        vectors["storage"].append({
            "acct": ak.hex(), "slot": sk.hex(),
            "present": sk in accounts[ak][3],
            "root": r.hex(), "nodes": [n.hex() for n in nodes],
        })
    out = os.path.join(os.path.dirname(__file__), "proof_vectors.json")
    json.dump(vectors, open(out, "w"), indent=1)
    print("wrote", out, len(vectors["accounts"]), "account +",
          len(vectors["storage"]), "storage vectors; root", root.hex()[:16])


if __name__ == "__main__":
    main()
