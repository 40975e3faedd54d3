import sys, os, ctypes
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from datafusion_ballista_amd import gpu
ctx = gpu.GpuStageContext(0)
dev = torch.device("cuda:0")
g = torch.Generator(device=dev); g.manual_seed(7)
nb, np_ = 3_000_000, 69_000_000
bk = torch.randperm(15_000_000, device=dev, dtype=torch.int64)[:nb] + 1
pk = torch.randint(1, 15_000_001, (np_,), generator=g, device=dev, dtype=torch.int64)
bk_h = bk.cpu().numpy(); pk_h = pk.cpu().numpy()
present = np.zeros(15_000_001, dtype=bool); present[bk_h] = True
want = int(present[pk_h].sum())
print("expected matches:", want, flush=True)
def col(t):
    return gpu.BgColumn(gpu.BG_DT_INT64, 0, 0, 0, t.data_ptr(), None, None, t.shape[0])
j = gpu.GpuHashJoin(ctx, col(bk), nb)
for r in range(3):
    pp, bb, m = j.probe(col(pk), np_)
    print("same-build probe", r, "matches", m, flush=True)
j.free()
for r in range(2):
    j2 = gpu.GpuHashJoin(ctx, col(bk), nb)
    pp, bb, m = j2.probe(col(pk), np_)
    print("rebuild", r, "matches", m, flush=True)
    j2.free()
