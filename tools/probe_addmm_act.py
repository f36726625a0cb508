import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def t(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e6

B, K, N = 65536, 512, 1024
x = torch.randn(B, K, device="cuda").bfloat16()
w = torch.randn(N, K, device="cuda").bfloat16()
b = torch.randn(N, device="cuda").bfloat16()
print("addmm        :", f"{t(lambda: torch.addmm(b, x, w.t())):7.1f} us")
try:
    out = torch._addmm_activation(b, x, w.t(), use_gelu=False)
    ref = torch.relu(torch.addmm(b, x, w.t()).float())
    err = (out.float() - ref).abs().max()
    print("addmm_act ok, max err", float(err))
    print("addmm_act    :", f"{t(lambda: torch._addmm_activation(b, x, w.t(), use_gelu=False)):7.1f} us")
except Exception as e:
    print("addmm_activation unavailable:", type(e).__name__, e)
print("addmm+relu_  :", f"{t(lambda: torch.relu_(torch.addmm(b, x, w.t()))):7.1f} us")
# bias grad reduce baseline
g = torch.randn(B, N, device="cuda").bfloat16()
print("bias sum0    :", f"{t(lambda: g.sum(0)):7.1f} us")
print("bias sum0 f32:", f"{t(lambda: g.float().sum(0)):7.1f} us")
