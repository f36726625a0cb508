import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from distributed_embeddings_amd.ops import _backend
ext = _backend.ops()

def timeit(fn, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

rows, width = 188_000_000, 128   # 96 GB fp32 — the DLRM fused table size
w = torch.empty(rows, width, device="cuda")
w.normal_()  # touch all pages
nids = 213_000
pool_warm = torch.randint(0, rows, (nids,), device="cuda")
splits = torch.arange(nids + 1, device="cuda")
pools = [torch.randint(0, rows, (nids,), device="cuda") for _ in range(16)]
it = [0]
def cold_ids():
    it[0] += 1
    return pools[it[0] % len(pools)]

us = timeit(lambda: ext.csr_lookup_forward(w, pool_warm, splits, False))
print(f"fused warm-ids   {us:9.1f} us  ({nids*width*4/us/1e3:.0f} GB/s read)")
us = timeit(lambda: ext.csr_lookup_forward(w, cold_ids(), splits, False))
print(f"fused cold-ids   {us:9.1f} us  ({nids*width*4/us/1e3:.0f} GB/s read)")
us = timeit(lambda: torch.index_select(w, 0, cold_ids()))
print(f"index_select cold{us:9.1f} us")
# power-law cold
from distributed_embeddings_amd.utils.input_gen import power_law_ids
ppools = [power_law_ids(rows, (nids,), 1.05).cuda() for _ in range(16)]
def pcold():
    it[0] += 1
    return ppools[it[0] % len(ppools)]
us = timeit(lambda: ext.csr_lookup_forward(w, pcold(), splits, False))
print(f"fused powerlaw   {us:9.1f} us")
# sorted cold ids (locality)
spool = [torch.sort(p)[0].contiguous() for p in pools]
def scold():
    it[0] += 1
    return spool[it[0] % len(spool)]
us = timeit(lambda: ext.csr_lookup_forward(w, scold(), splits, False))
print(f"fused sorted-ids {us:9.1f} us")
