import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from distributed_embeddings_amd.ops import _backend
ext = _backend.ops()

def timeit(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us

vocab, width, b = 39_000_000, 128, 8192
w = torch.randn(vocab, width, device="cuda")
ids = torch.randint(0, vocab, (b,), device="cuda")
splits = torch.arange(b + 1, device="cuda")

us = timeit(lambda: ext.csr_lookup_forward(w, ids, splits, False))
print(f"csr_fwd hot1 random {us:8.1f} us  ({b*width*4*2/us/1e3:.0f} GB/s eff)")
us = timeit(lambda: torch.index_select(w, 0, ids))
print(f"index_select        {us:8.1f} us")

# power-law ids (hot rows)
from distributed_embeddings_amd.utils.input_gen import power_law_ids
pids = power_law_ids(vocab, (b,), 1.05).cuda()
us = timeit(lambda: ext.csr_lookup_forward(w, pids, splits, False))
print(f"csr_fwd powerlaw    {us:8.1f} us")

# hotness 8
ids8 = torch.randint(0, vocab, (b*8,), device="cuda")
splits8 = torch.arange(0, b*8+1, 8, device="cuda")
us = timeit(lambda: ext.csr_lookup_forward(w, ids8, splits8, False))
print(f"csr_fwd hot8        {us:8.1f} us  ({b*8*width*4/us/1e3:.0f} GB/s read)")

# big batch: 64k x hot1
ids64k = torch.randint(0, vocab, (65536,), device="cuda")
sp64k = torch.arange(65536 + 1, device="cuda")
us = timeit(lambda: ext.csr_lookup_forward(w, ids64k, sp64k, False))
print(f"csr_fwd hot1 64k    {us:8.1f} us")
us = timeit(lambda: torch.index_select(w, 0, ids64k))
print(f"index_select 64k    {us:8.1f} us")

# backward pipeline
g = torch.randn(b, width, device="cuda")
us = timeit(lambda: ext.csr_lookup_backward(g, ids, splits, vocab, False))
print(f"csr_bwd 8k          {us:8.1f} us")

# narrow width
w16 = torch.randn(1_000_000, 16, device="cuda")
ids16 = torch.randint(0, 1_000_000, (b,), device="cuda")
us = timeit(lambda: ext.csr_lookup_forward(w16, ids16, splits, False))
print(f"csr_fwd w16 hot1    {us:8.1f} us")

# long-segment (tiny vocab) backward — the split-kernel stress
w3 = torch.randn(3, width, device="cuda")
ids3 = torch.randint(0, 3, (8192,), device="cuda")
us = timeit(lambda: ext.csr_lookup_backward(g, ids3, splits, 3, False))
print(f"csr_bwd vocab3      {us:8.1f} us")
