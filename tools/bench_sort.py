import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from distributed_embeddings_amd import Ragged, embedding_lookup

def timeit(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

# correctness: custom sort via full backward vs CPU oracle
torch.manual_seed(0)
vocab = 188_000_000
w = torch.randn(2000, 64).cuda().requires_grad_(True)
ids = torch.randint(0, 2000, (50000,)).cuda()
splits = torch.arange(0, 50001, 5, device="cuda")
out = embedding_lookup(w, Ragged(ids, splits), "mean")
out.sum().backward()
g = w.grad.coalesce()
w2 = w.detach().cpu().clone().requires_grad_(True)
out2 = embedding_lookup(w2, Ragged(ids.cpu(), splits.cpu()), "mean")
out2.sum().backward()
g2 = w2.grad.coalesce()
assert torch.equal(g.indices().cpu(), g2.indices()), "custom sort: wrong ids"
err = (g.values().cpu() - g2.values()).abs().max()
print(f"custom-sort backward correct, max err {float(err):.2e}")

# perf: backward timing custom vs rocprim (env toggles at import... need subprocess)
gr = torch.randn(10000, 64, device="cuda")
big_ids = torch.randint(0, vocab, (213000,), device="cuda")
big_splits = torch.arange(0, 213001, 22, device="cuda")[:10001].contiguous()
big_splits[-1] = 213000
from distributed_embeddings_amd.ops import _backend
ext = _backend.ops()
us = timeit(lambda: ext.csr_lookup_backward(gr, big_ids[:int(big_splits[-1])], big_splits, vocab, False))
print(f"backward 213k ids vocab=188M: {us:.0f} us  (sort={'rocprim' if os.environ.get('DE_USE_ROCPRIM_SORT')=='1' else 'custom'})")
