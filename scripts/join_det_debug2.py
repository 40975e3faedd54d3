import sys, os, ctypes
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from datafusion_ballista_amd import gpu
ctx = gpu.GpuStageContext(0)
dev = torch.device("cuda:0")
nb = 3_000_000
g = torch.Generator(device=dev); g.manual_seed(11)
bk = (torch.randperm(15_000_000, device=dev, dtype=torch.int64,
                     generator=g)[:nb] + 1).contiguous()
bk_h = bk.cpu().numpy()
def col(t):
    return gpu.BgColumn(gpu.BG_DT_INT64, 0, 0, 0, t.data_ptr(), None, None, t.shape[0])
j = gpu.GpuHashJoin(ctx, col(bk), nb)
ctx.synchronize()
# reach into the handle: BgJoinTable layout {nodes, head, n_build, mask, ...}
class T(ctypes.Structure):
    _fields_ = [("nodes", ctypes.c_void_p), ("head", ctypes.c_void_p),
                ("n_build", ctypes.c_int64), ("mask", ctypes.c_uint64)]
t = ctypes.cast(j._handle, ctypes.POINTER(T)).contents
nbuk = t.mask + 1
print("nb buckets", nbuk, "nodes@", hex(t.nodes), "head@", hex(t.head))
head = np.empty(nbuk, dtype=np.int32)
nodes = np.empty(nb * 2, dtype=np.uint64)
gpu._check(ctx.L.bg_memcpy_d2h(head.ctypes.data_as(ctypes.c_void_p),
           ctypes.c_void_p(t.head), ctypes.c_uint64(4 * nbuk)), "d2h")
gpu._check(ctx.L.bg_memcpy_d2h(nodes.ctypes.data_as(ctypes.c_void_p),
           ctypes.c_void_p(t.nodes), ctypes.c_uint64(16 * nb)), "d2h")
# oracle-side hash (bit-exact with device)
import oracle
hh = oracle.hash_columns([("i64", bk_h)], nb)
buck = (hh & t.mask).astype(np.int64)
# walk every chain, collect reachable node ids
reach = np.zeros(nb, dtype=bool)
bad_next = 0
for b in np.unique(buck):
    cur = head[b]
    steps = 0
    while cur >= 0 and steps <= nb:
        if cur >= nb: bad_next += 1; break
        reach[cur] = True
        cur = int(nodes[2*cur+1] & 0xFFFFFFFF) - 1
        steps += 1
lost = np.where(~reach)[0]
print("unreachable nodes:", len(lost), "bad_next:", bad_next)
if len(lost):
    i = int(lost[0])
    b = buck[i]
    print("example lost node", i, "key", bk_h[i], "bucket", b, "head[b]", head[i and 0] if False else head[b])
    # chain at that bucket
    cur = head[b]; chain=[]
    while cur >= 0 and len(chain) < 10:
        chain.append((cur, int(nodes[2*cur]), int(nodes[2*cur+1]>>32), int(nodes[2*cur+1]&0xffffffff)-1))
        cur = int(nodes[2*cur+1] & 0xFFFFFFFF) - 1
    print("chain(id,key,orig,next):", chain)
    print("lost node record: key", int(nodes[2*i]), "orig", int(nodes[2*i+1]>>32), "next", int(nodes[2*i+1]&0xffffffff)-1)
    # which nodes share bucket b?
    members = np.where(buck == b)[0]
    print("bucket members:", members, [int(bk_h[m]) for m in members])
j.free()
