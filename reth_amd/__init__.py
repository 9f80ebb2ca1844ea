"""reth_amd — MI355X-native state-root engine for reth's MPT commitment path.

Product surface:
  - reth_amd.engine: ctypes binding over the C-ABI of libsre (include/sre.h),
    the HIP/CDNA4 engine. GPU-only; raises if the extension is missing.
  - reth_amd.gen: deterministic synthetic-state generator.
  - reth_amd.sharding: multi-GPU partition + subtrie-digest exchange
    (torch.distributed over RCCL) and top-of-trie finish.

The CPU oracle lives OUTSIDE this package (oracle/) and is test
infrastructure only.
"""
