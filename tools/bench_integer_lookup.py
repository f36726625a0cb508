import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from distributed_embeddings_amd import IntegerLookup

def timeit(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

torch.manual_seed(0)
# build phase: fresh keys (all inserts)
lk = IntegerLookup(max_tokens=40_000_000).cuda()
keys_new = (torch.randperm(30_000_000, device="cuda")[:8_000_000] * 2654435761) & ((1 << 62) - 1)
t0 = time.perf_counter(); out = lk(keys_new); torch.cuda.synchronize()
build_ms = (time.perf_counter() - t0) * 1000
print(f"vocab build: 8M fresh int64 keys in {build_ms:.1f} ms "
      f"({8_000_000/build_ms*1000/1e6:.0f}M inserts/s)")

# steady-state lookup: existing keys, batch 64k x 26-feature equivalent
batch = 65536 * 26
idx = torch.randint(0, keys_new.numel(), (batch,), device="cuda")
keys_seen = keys_new[idx]
us = timeit(lambda: lk(keys_seen))
print(f"lookup {batch} existing keys: {us:.0f} us ({batch/us:.0f}M lookups/s)")
