import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributed_embeddings_amd.ops import _backend
from distributed_embeddings_amd.ops.dot_interact import _DotInteract
ext = _backend.ops()
torch.manual_seed(0)
# lookup kernels
w = torch.randn(1_000_000, 128, device="cuda")
ids = torch.randint(0, 1_000_000, (200_000,), device="cuda")
splits = torch.arange(0, 200_001, 20, device="cuda")
for _ in range(3):
    out = ext.csr_lookup_forward(w, ids, splits, False)
g = torch.randn(10_000, 128, device="cuda")
for _ in range(3):
    ext.csr_lookup_backward(g, ids, splits, 1_000_000, False)
# dot interact MFMA
feats = torch.randn(4096, 27, 128).bfloat16().cuda().requires_grad_(True)
for _ in range(3):
    o = _DotInteract.apply(feats, 512)
    o.backward(torch.randn_like(o))
torch.cuda.synchronize()
print("pmc probe done")
