#!/usr/bin/env python3
"""Round-2 robustness soak: repeated build/solve/teardown cycles over the
round-2 code paths (skew-split CSR shards, fused bf16-K Gram trials,
checkpoint/resume with margin tracking), with memory watermarks printed so
leaks or allocator growth are visible. Bounded: fixed shard sizes, fixed
cycle count, suitable for a ~3-minute gpurun slot."""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from sparkagd_amd import (  # noqa: E402
    LogisticGradient,
    MultinomialLogisticGradient,
    SquaredL2Updater,
    run,
)
from sparkagd_amd.data import (  # noqa: E402
    generate_csr_problem,
    generate_dense_problem,
    generate_multiclass_problem,
)

CYCLES = int(os.environ.get("SOAK_CYCLES", "12"))


def mem():
    return (torch.cuda.memory_allocated() / 2**30,
            torch.cuda.max_memory_allocated() / 2**30)


def main() -> int:
    assert torch.cuda.is_available()
    dev = "cuda:0"
    t0 = time.perf_counter()
    losses = []
    for cyc in range(CYCLES):
        # 1) zipf CSR (heavy split structures built + torn down each cycle)
        shard, _ = generate_csr_problem(200_000, 2_000_000, 32,
                                        seed=100 + cyc, device=dev,
                                        col_dist="zipf", zipf_a=1.1)
        w0 = torch.zeros(2_000_000, device=dev, dtype=torch.float32)
        w, h = run(shard, LogisticGradient(), SquaredL2Updater(), 1e-10, 12,
                   1e-3, w0, 1.0, math.inf, 0.5, 0.9, True,
                   loss_history_mode="backtrack")
        assert h[-1] < h[0] and math.isfinite(h[-1])
        losses.append(h[-1])
        del shard, w, w0

        # 2) fused bf16-K gram + checkpoint/resume on the direct path
        dshard, _ = generate_dense_problem(8192, 200_000, seed=200 + cyc,
                                           device=dev, dtype=torch.bfloat16)
        w0 = torch.zeros(200_000, device=dev, dtype=torch.float32)
        wg, hg = run(dshard, LogisticGradient(), SquaredL2Updater(), 1e-10,
                     25, 1e-3, w0, 1.0, math.inf, 0.5, 0.9, True,
                     solver="gram", loss_history_mode="backtrack")
        assert math.isfinite(hg[-1]) and hg[-1] < hg[0]
        ck = f"/tmp/soak_{cyc}.safetensors"
        run(dshard, LogisticGradient(), SquaredL2Updater(), 0.0, 6, 1e-3, w0,
            1.0, math.inf, 0.5, 0.9, True, checkpoint_path=ck,
            checkpoint_every=3, track_margins=False)
        w_r, h_r = run(dshard, LogisticGradient(), SquaredL2Updater(), 0.0,
                       12, 1e-3, w0, 1.0, math.inf, 0.5, 0.9, True,
                       resume_from=ck, track_margins=False)
        assert len(h_r) == 12 and math.isfinite(h_r[-1])
        os.unlink(ck)
        del dshard, wg, w_r, w0

        # 3) multiclass gram (fused bookkeeping, padded columns)
        mshard, _ = generate_multiclass_problem(4096, 65536, 8,
                                                seed=300 + cyc, device=dev,
                                                dtype=torch.bfloat16,
                                                label_noise=0.3)
        w0 = torch.zeros(65536 * 8, device=dev, dtype=torch.float32)
        wm, hm = run(mshard, MultinomialLogisticGradient(8),
                     SquaredL2Updater(), 1e-10, 10, 1e-2, w0,
                     1.0, math.inf, 0.5, 0.9, True, solver="gram",
                     loss_history_mode="backtrack")
        assert math.isfinite(hm[-1]) and hm[-1] < hm[0]
        del mshard, wm, w0

        alloc, peak = mem()
        print(f"cycle {cyc + 1}/{CYCLES}: csr_loss={losses[-1]:.5f} "
              f"alloc={alloc:.2f} GiB peak={peak:.2f} GiB "
              f"t={time.perf_counter() - t0:.1f}s", flush=True)

    # losses across cycles must be in a sane band (different seeds)
    spread = max(losses) - min(losses)
    print(f"SOAK_OK cycles={CYCLES} csr_loss_spread={spread:.4f} "
          f"wall={time.perf_counter() - t0:.1f}s", flush=True)
    return 0


if __name__ == "__main__":
    sys.exit(main())
