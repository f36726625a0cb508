import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from distributed_embeddings_amd.ops import _backend
ext = _backend.ops()
def timeit(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6
torch.manual_seed(0)
for nnz, vocab, label in [(213_000, 188_000_000, "dlrm"), (2_000_000, 4_000_000, "small-grp")]:
    gr = torch.randn(10000, 64, device="cuda")
    ids = torch.randint(0, vocab, (nnz,), device="cuda")
    splits = torch.linspace(0, nnz, 10001, device="cuda").long()
    us = timeit(lambda: ext.csr_lookup_backward(gr, ids, splits, vocab, False))
    print(f"{label}: backward {nnz} ids: {us:.0f} us")
