import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
from lightctr_amd.data.synthetic import SyntheticCriteo
from lightctr_amd.ops._extension import require_hip_ops

ops = require_hip_ops()
F, nf, K = 1 << 24, 39, 8
gen = SyntheticCriteo(num_features=F, seed=5, device="cuda")
row_ptr, fields, fids, vals, labels = gen.batch(65536)
g = torch.Generator().manual_seed(1)
W = torch.zeros(F, device="cuda")
V = (torch.randn(F, nf, K, generator=g) * 0.01).cuda()

def t(fn, reps=15):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps * 1e6

p1 = ops.ffm_forward(row_ptr, fields, fids, vals, W, V)
p2 = ops.ffm_forward_pp(row_ptr, fields, fids, vals, W, V)
print("maxdiff", (p1 - p2).abs().max().item())
print(f"group-layout fwd {t(lambda: ops.ffm_forward(row_ptr, fields, fids, vals, W, V)):8.1f} us")
print(f"lane-per-pair    {t(lambda: ops.ffm_forward_pp(row_ptr, fields, fids, vals, W, V)):8.1f} us")
