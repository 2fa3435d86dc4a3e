"""300-iteration multiclass soak: tracked (GEMM margins + axpby propagation)
vs untracked trajectories — bounds the bf16-GEMM margin-tracking drift."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math, torch
from sparkagd_amd import MultinomialLogisticGradient, SquaredL2Updater, run
from sparkagd_amd.data import generate_multiclass_problem

dev = "cuda" if torch.cuda.is_available() else "cpu"
shard, _ = generate_multiclass_problem(50000, 2048, 16, seed=71, device=dev,
                                       dtype=torch.bfloat16 if dev == "cuda" else torch.float64,
                                       label_noise=0.2)
K = 16
w0 = torch.zeros(2048 * K, device=dev,
                 dtype=torch.float32 if dev == "cuda" else torch.float64)
args = (shard, MultinomialLogisticGradient(K), SquaredL2Updater(), 0.0, 300,
        1e-3, w0, 1.0, math.inf, 0.5, 0.9, True)
w_t, h_t = run(*args, loss_history_mode="backtrack", track_margins=True)
w_u, h_u = run(*args, loss_history_mode="backtrack", track_margins=False)
n = min(len(h_t), len(h_u))
mx = max(abs(a - b) / max(abs(b), 1e-12) for a, b in zip(h_t[:n], h_u[:n]))
wdrift = float(torch.norm(w_t - w_u) / (torch.norm(w_u) + 1e-30))
print(f"iters tracked={len(h_t)} untracked={len(h_u)}  "
      f"final {h_t[-1]:.8f} vs {h_u[-1]:.8f}  max_rel_loss_diff={mx:.3e}  "
      f"w_rel_diff={wdrift:.3e}")
assert all(math.isfinite(x) for x in h_t)
