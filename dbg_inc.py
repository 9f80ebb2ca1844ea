import torch, numpy as np, sys
sys.path.insert(0, ".")
from reth_amd import gen
from reth_amd.engine import StateRootEngine, DELTA_DTYPE, STORAGE_DTYPE

def mem(tag):
    free, total = torch.cuda.mem_get_info()
    print(f"{tag}: free {free/2**30:.1f} GiB / {total/2**30:.1f}", flush=True)

torch.cuda.init(); mem("start")
eng = StateRootEngine(0)
acct_t, st_t = gen.gen_state_torch(10_000_000, 64,
                                   eng.keccak_batch_device, device="cuda:0")
mem("generated")
eng.set_device_tensors(acct_t, st_t)
torch.cuda.empty_cache(); mem("set+empty_cache")
r = eng.root_retaining(); mem("after retaining")
# tiny idempotent delta
d = np.zeros(1, dtype=DELTA_DTYPE)
d[0]["key"][:] = 1; d[0]["nonce"] = 1
d[0]["balance"][31] = 1
s = np.zeros(0, dtype=STORAGE_DTYPE)
try:
    r2 = eng.incremental_root(d, s); mem("after inc step1")
    r3 = eng.incremental_root(d, s); mem("after inc step2")
    print("ok", r2 == r3)
except RuntimeError as e:
    mem("at failure"); print("ERR", e)
