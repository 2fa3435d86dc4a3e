import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time, torch
from sparkagd_amd.data import generate_multiclass_csr_problem
from sparkagd_amd.ops import hiplib, multiclass as mc

dev = torch.device("cuda")
shard, _ = generate_multiclass_csr_problem(1000000, 10000000, 64, 16, seed=41, device=dev)
k, kc = 16, 16
W = (torch.randn(shard.d * k, device=dev) / 8).contiguous()

def timeit(fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): out = fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps, out

t_mine, z1 = timeit(lambda: hiplib.csr_margins_multi(shard.rowptr, shard.col, shard.val, W, k, kc, shard.d))
A = torch.sparse_csr_tensor(shard.rowptr.to(torch.int64), shard.col.to(torch.int64), shard.val, size=(shard.n, shard.d))
W2 = W.reshape(shard.d, k)
t_lib, z2 = timeit(lambda: A @ W2)
print(f"gather kernel: {t_mine*1e3:.2f} ms   rocsparse spmm: {t_lib*1e3:.2f} ms")
err = float(torch.norm(z1.reshape(-1, kc)[:, :k] - z2) / torch.norm(z2))
print("rel err", err)
