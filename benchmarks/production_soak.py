"""Production soak: continuous bench-scale training with periodic
checkpoints and a hard mid-run SIGKILL, then resume-from-checkpoint and
loss-continuity verification (SURVEY.md §5 failure detection / recovery).
"""
import json
import math
import os
import signal
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

CKPT = "gpurun_out/soak_ckpt.safetensors"
METRICS = "gpurun_out/soak_metrics.jsonl"

TRAIN = r"""
import math, sys, torch
sys.path.insert(0, %(repo)r)
from sparkagd_amd import LogisticGradient, SquaredL2Updater, run, generate_dense_problem
from sparkagd_amd.utils.metrics import JsonlMetrics
dev = "cuda" if torch.cuda.is_available() else "cpu"
shard, _ = generate_dense_problem(16384, 1_000_000, seed=99, device=dev,
                                  dtype=torch.bfloat16 if dev == "cuda" else torch.float64)
w0 = torch.zeros(1_000_000, device=dev,
                 dtype=torch.float32 if dev == "cuda" else torch.float64)
kw = dict(loss_history_mode="backtrack", checkpoint_path=%(ckpt)r,
          checkpoint_every=500, metrics=JsonlMetrics(%(metrics)r))
%(resume)s
w, h = run(shard, LogisticGradient(), SquaredL2Updater(), 0.0, %(iters)d,
           1e-3, w0, 1.0, math.inf, 0.5, 0.9, True, **kw)
print("DONE", len(h), h[-1], flush=True)
"""

os.makedirs("gpurun_out", exist_ok=True)
for f in (CKPT, METRICS):
    if os.path.exists(f):
        os.unlink(f)

# Phase 1: train with periodic checkpoints; hard-kill after ~100 s
code = TRAIN % {"repo": REPO, "ckpt": CKPT, "metrics": METRICS,
                "iters": 1_000_000, "resume": ""}
p = subprocess.Popen([sys.executable, "-c", code])
time.sleep(100)
if p.poll() is None:
    p.send_signal(signal.SIGKILL)  # exact PID of our own child
    p.wait()
    print("phase1: killed mid-run (as intended)")
else:
    print(f"phase1: exited early rc={p.returncode}")
assert os.path.exists(CKPT), "no checkpoint was written before the kill"

lines = [json.loads(x) for x in open(METRICS)]
iters_done = lines[-1]["iter"]
assert all(math.isfinite(r["loss"]) for r in lines)
bt = sum(r["n_backtracks"] for r in lines)
rs = sum(1 for r in lines if r["restarted"])
print(f"phase1: {iters_done} iterations before kill, all losses finite, "
      f"{bt} backtracks, {rs} restarts, last loss {lines[-1]['loss']:.8f}")

# Phase 2: resume from the checkpoint, run 500 more iterations
from sparkagd_amd.utils.checkpoint import load_checkpoint
ck = load_checkpoint(CKPT)
ck_iter, ck_loss = ck["iter"], ck["loss_history"][-1]
code2 = TRAIN % {"repo": REPO, "ckpt": CKPT, "metrics": METRICS,
                 "iters": ck_iter + 500,
                 "resume": f"kw['resume_from'] = {CKPT!r}"}
out = subprocess.run([sys.executable, "-c", code2], capture_output=True,
                     text=True, timeout=400)
assert out.returncode == 0, out.stderr[-2000:]
done = [ln for ln in out.stdout.splitlines() if ln.startswith("DONE")][0]
_, n_hist, final_loss = done.split()
final_loss = float(final_loss)
print(f"phase2: resumed at iter {ck_iter} (ckpt loss {ck_loss:.8f}), "
      f"ran to {n_hist} total, final loss {final_loss:.8f}")
assert final_loss <= ck_loss + 1e-9, "loss regressed after resume"
print("SOAK OK")
