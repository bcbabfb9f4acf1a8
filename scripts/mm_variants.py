import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from collections import Counter
import torch

dev = "cuda:0"
w = torch.randn(64, 192, device=dev, dtype=torch.bfloat16)
x = torch.randn(2, 192, 8192, device=dev, dtype=torch.bfloat16)

def clones(label, fn):
    fn(); torch.cuda.synchronize()
    with torch.profiler.profile(record_shapes=True) as p:
        y = fn()
        torch.cuda.synchronize()
    c = Counter(e.name for e in p.events())
    import time
    t0 = time.perf_counter()
    for _ in range(50):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 50 * 1e6
    print(f"{label:40s} clone={c['aten::clone']} copy_={c['aten::copy_']} {dt:8.1f} us")
    return y

a = clones("matmul(2D,3D)", lambda: torch.matmul(w, x))
b = clones("matmul(w.unsqueeze(0), x)", lambda: torch.matmul(w.unsqueeze(0), x))
c = clones("bmm(w.expand, x)", lambda: torch.bmm(w.unsqueeze(0).expand(x.shape[0], -1, -1), x))
d = clones("per-batch mm loop", lambda: torch.stack([torch.mm(w, x[i]) for i in range(x.shape[0])]))
print("equal:", torch.allclose(a, b), torch.allclose(a, c), torch.allclose(a, d, atol=1e-2))
# transposed-A variant for dx
wt = w.t()
clones("matmul(w.t().unsqueeze(0), y3d)", lambda: torch.matmul(wt.unsqueeze(0), a))
clones("bmm(w.t().expand, y3d)", lambda: torch.bmm(wt.unsqueeze(0).expand(2, -1, -1), a))

print("--- context probes ---")
class F1(torch.autograd.Function):
    @staticmethod
    def forward(ctx, w_, x_):
        return torch.matmul(w_, x_)
    @staticmethod
    def backward(ctx, g):
        return None, None

clones("inside Function", lambda: F1.apply(w, x))
def under_ac():
    with torch.autocast("cuda", dtype=torch.bfloat16):
        return torch.matmul(w, x)
clones("under autocast", under_ac)
def both():
    with torch.autocast("cuda", dtype=torch.bfloat16):
        return F1.apply(w, x)
clones("Function under autocast", both)
xg = x.clone().requires_grad_(True)
clones("matmul requires_grad", lambda: torch.matmul(w, xg))
wg = w.clone().requires_grad_(True)
clones("matmul both grad", lambda: torch.matmul(wg, xg))
