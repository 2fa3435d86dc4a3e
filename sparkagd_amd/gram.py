"""Dual-space (Gram) AGD solver for the n ≪ d regime.

Margin-state tracking (optimizer.py) eliminates the A·y / A·x passes; this
module takes the idea to its conclusion. Every AT iterate lives in
span{x0, g_1, g_2, ...} where g_t = Aᵀm_t / c, so with the Gram operator
K = A_local · A_globalᵀ precomputed ONCE (a plain GEMM — the one true GEMM
in this workload, built with rocBLAS f32 via torch.matmul in d-chunks):

* the new basis vector's margins  A·g_t = K·m_global / c   — an O(n_local ·
  n_global) GEMV instead of TWO O(n·d) shard passes;
* every weight-space scalar (norms, dots for backtracking / convergence /
  restart) is a quadratic form over the small fp64 basis Gram matrix
  G[i][j] = v_i · v_j, maintained incrementally from margin-space dots
  (v_t·v_j = Σ_ranks m_t·gm_j / c, all-reduced in fp64);
* x is materialized only at the end: x = cx₀·x0 + Aᵀ(Σ_j cx_j m_j)/c —
  one transpose pass.

Per-iteration cost drops from O(n·d) to O(n_local·n_global): at the
headline config (d=10⁶, n=16384/GPU) that is ~1 GB of K-traffic per trial
instead of ~66 GB of shard traffic. The trade-offs, stated plainly:

* one-time K build: n_local·n_global·d FLOPs (f32 GEMM) + n_global·d bytes
  of shard exchange across ranks, and n_local·n_global·4 B of HBM for K;
* per-GPU iteration work grows with world size (K row-block is
  n_local × n_global), so weak scaling of iteration *throughput* is flat —
  absolute time-to-solution still wins whenever n_global ≲ d;
* the gradient basis grows by one vector per backtracking trial, so the
  basis stores (Mstore f32 + XB f64: ~12·max_basis·n_local bytes with
  max_basis = 8·num_iterations+8) bound the horizon — fine for the
  hundreds-of-iterations solves this regime needs (157 MB at the headline
  config × 300 iterations); open-ended training belongs to the direct
  solver;
* requires a dense shard, an AFFINE prox (Simple/SquaredL2), full-batch
  evaluations (no mini-batch masks), and fp32 (or fp64) accumulation
  identical in class to the direct path. Multiclass (softmax) gradients are
  supported: margins/multipliers are padded [n·KC] flats whose zero pad
  columns leave every inner product unchanged, and K applies column-wise
  (one rocBLAS GEMM per trial).

The trajectory is the same mathematics as the direct solver up to fp
rounding (asserted by tests/test_gram.py against the direct path).
"""

from __future__ import annotations

import logging
import math
import time
from typing import List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from . import ops
from .data import DenseShard
from .parallel.comm import Communicator

logger = logging.getLogger(__name__)


class GramOperator:
    """K = A_local · A_globalᵀ, built chunked; matvec via the margins kernel
    (K is just a dense f32/f64 'shard' whose feature dimension is n_global)."""

    def __init__(self, shard: DenseShard, comm: Communicator,
                 chunk_rows: int = 1024, d_chunk: int = 65536,
                 mem_budget_bytes: int = 64 << 30,
                 k_dtype: Optional[str] = None):
        t0 = time.perf_counter()
        if k_dtype is None:  # env override for A/B runs (bench.py)
            import os

            k_dtype = os.environ.get("SPARKAGD_GRAM_K", "auto")
        self.comm = comm
        A = shard.features
        dev = A.device
        acc = torch.float64 if A.dtype == torch.float64 else torch.float32
        self.acc = acc
        # K storage dtype. The per-trial K·m GEMV streams the whole K at the
        # memory roofline (n_local*n_global*itemsize bytes), so storing K in
        # bf16 halves trial time — and matches the direct path's dtype
        # discipline exactly (the direct solver streams bf16 A with f32
        # accumulation; gm from a bf16 K carries the same class of rounding
        # as margins from bf16 features). The coefficient-space algebra
        # stays self-consistent: G is built from f64 dots of the margins
        # actually stored, and the final x materializes from exact
        # A^T(sum c_j m_j)/c. 'auto' = bf16 iff the shard is bf16 (GPU);
        # 'f32' forces full precision (and is the rule off-GPU/f64).
        if k_dtype not in ("auto", "bf16", "f32"):
            raise ValueError("k_dtype must be auto|bf16|f32")
        use_bf16_k = (A.dtype == torch.bfloat16 and A.is_cuda
                      and k_dtype in ("auto", "bf16"))
        self.k_dtype = torch.bfloat16 if use_bf16_k else acc
        n_local, d = A.shape
        counts = [n_local]
        if comm.world_size > 1:
            ct = torch.tensor([n_local], dtype=torch.int64,
                              device=dev if dist.get_backend() == "nccl" else "cpu")
            cl = [torch.zeros_like(ct) for _ in range(comm.world_size)]
            dist.all_gather(cl, ct)
            counts = [int(c) for c in cl]
        self.counts = counts
        self.offsets = np.concatenate([[0], np.cumsum(counts)])
        n_global = int(self.offsets[-1])
        self.n_local, self.n_global = n_local, n_global

        k_bytes = n_local * n_global * self.k_dtype.itemsize
        if k_bytes > mem_budget_bytes:
            raise MemoryError(
                f"Gram matrix needs {k_bytes/2**30:.1f} GiB > budget "
                f"{mem_budget_bytes/2**30:.1f} GiB — use the direct solver")

        # bf16 shards on GPU: one hipBLASLt bf16-in/f32-out GEMM per block
        # (fp32 accumulation; ~15x the chunked f32 rocBLAS route, no cast
        # traffic). Other dtypes/devices: chunked f32/f64 torch GEMMs.
        use_lt = A.dtype == torch.bfloat16 and A.is_cuda
        K = torch.empty((n_local, n_global), dtype=self.k_dtype, device=dev)
        if use_lt and comm.world_size == 1 and self.k_dtype != torch.bfloat16:
            from .ops.hiplib import gemm_bf16f32_nt

            gemm_bf16f32_nt(A, A, K)  # one bf16->f32 GEMM, C written in place
            self.K = K
            self.build_seconds = time.perf_counter() - t0
            return
        for r in range(comm.world_size):
            r_lo = int(self.offsets[r])
            n_r = counts[r]
            for c_lo in range(0, n_r, chunk_rows):
                c_hi = min(c_lo + chunk_rows, n_r)
                if comm.world_size > 1:
                    buf = torch.empty((c_hi - c_lo, d), dtype=A.dtype, device=dev)
                    if comm.rank == r:
                        buf.copy_(A[c_lo:c_hi])
                    dist.broadcast(buf, src=r)
                else:
                    buf = A[c_lo:c_hi]
                dst = K[:, r_lo + c_lo: r_lo + c_hi]
                if use_lt:
                    from .ops.hiplib import gemm_bf16f32_nt

                    block = torch.empty((n_local, c_hi - c_lo), dtype=torch.float32,
                                        device=dev)
                    gemm_bf16f32_nt(A, buf.contiguous(), block)
                    dst.copy_(block)  # f32 accumulation; rounded here iff K is bf16
                    del block
                else:
                    tmp = torch.zeros((n_local, c_hi - c_lo), dtype=acc, device=dev)
                    for d_lo in range(0, d, d_chunk):
                        d_hi = min(d_lo + d_chunk, d)
                        tmp.addmm_(A[:, d_lo:d_hi].to(acc), buf[:, d_lo:d_hi].to(acc).T)
                    dst.copy_(tmp)
                    del tmp
                del buf
        self.K = K.contiguous()
        self.build_seconds = time.perf_counter() - t0

    def matvec(self, m_global: torch.Tensor, ncols: int = 1) -> torch.Tensor:
        """K @ m_global -> local margin slice (flat). ncols > 1 (multiclass:
        padded class columns) runs one GEMM instead of the GEMV margins
        kernel; zero pad columns stay exactly zero (bf16 rounding of an
        exact zero is an exact zero)."""
        if ncols > 1:
            mg = m_global.to(self.acc).reshape(self.n_global, ncols)
            if self.K.dtype == torch.bfloat16:
                from .ops.hiplib import gemm_bf16f32_nt

                z = torch.empty((self.n_local, ncols), dtype=torch.float32,
                                device=self.K.device)
                gemm_bf16f32_nt(self.K, mg.T.to(torch.bfloat16).contiguous(), z)
                return z.reshape(-1)
            return (self.K @ mg).reshape(-1)
        # the margins kernel streams bf16/f32/f64 K with f32/f64 m
        return ops.dense_margins(self.K, m_global.to(self.acc))

    def all_gather_m(self, m_local: torch.Tensor, ncols: int = 1) -> torch.Tensor:
        if self.comm.world_size == 1:
            return m_local
        # NCCL all_gather requires equal lengths: pad to the max count.
        mx = max(self.counts) * ncols
        buf = m_local
        if m_local.numel() != mx:
            buf = torch.zeros(mx, dtype=m_local.dtype, device=m_local.device)
            buf[: m_local.numel()] = m_local
        parts = [torch.empty(mx, dtype=m_local.dtype, device=m_local.device)
                 for _ in self.counts]
        dist.all_gather(parts, buf.contiguous())
        return torch.cat([p[: c * ncols] for p, c in zip(parts, self.counts)])


def run_gram(
    data: DenseShard,
    gradient,
    updater,
    convergence_tol: float,
    num_iterations: int,
    reg_param: float,
    initial_weights: torch.Tensor,
    L0: float = 1.0,
    Lexact: float = math.inf,
    beta: float = 0.5,
    alpha: float = 0.9,
    may_restart: bool = True,
    *,
    loss_history_mode: str = "backtrack",
    comm: Optional[Communicator] = None,
    metrics=None,
    iteration_hook=None,
    gram_op: Optional[GramOperator] = None,
    backtrack_tol: float = 1e-10,
) -> Tuple[torch.Tensor, List[float]]:
    """AGD in coefficient space over the gradient basis (same 12-parameter
    surface as optimizer.run; same AT/backtracking/restart state machine;
    returns (weights, loss_history))."""
    comm = comm or Communicator()
    if not getattr(updater, "AFFINE_PROX", False):
        raise ValueError("run_gram requires an affine prox updater (Simple/SquaredL2)")
    if getattr(data, "kind", None) != "dense":
        raise ValueError("run_gram requires a DenseShard")
    # Multiclass: margins/multipliers are padded [n*KC] flats; every piece of
    # the coefficient-space machinery below is shape-agnostic over the flat
    # length (pad columns are exactly zero everywhere so dots are unaffected),
    # and K applies column-wise (matvec ncols). The basis Gram matrix G holds
    # Frobenius inner products of the [d,K] basis matrices.
    ncols = 1
    if getattr(gradient, "IS_MULTICLASS", False):
        from .ops.multiclass import padded_k

        ncols = padded_k(gradient.num_classes)

    op = gram_op or GramOperator(data, comm)
    acc = op.acc
    dev = data.device
    n_local = data.n
    feats = data.features

    # count c (full batch, constant; sum of weights for weighted shards)
    if getattr(data, "sample_weight", None) is not None:
        cvec = data.sample_weight.to(torch.float64).sum().reshape(1).clone()
    else:
        cvec = torch.tensor([float(n_local)], dtype=torch.float64, device=dev)
    comm.allreduce_(cvec)
    c = float(cvec[0])

    x0 = initial_weights.clone()
    x0_nonzero = bool(torch.any(x0 != 0))
    xm0 = (gradient.margins(data, x0.to(acc)) if x0_nonzero
           else torch.zeros(n_local * ncols, dtype=acc, device=dev))
    norm_x0_sq = float((x0.to(torch.float64) ** 2).sum())

    # basis storage (index 0 = x0; gradients at 1..T)
    max_basis = 8 * num_iterations + 8
    G = np.zeros((max_basis + 1, max_basis + 1))
    G[0, 0] = norm_x0_sq
    flat_n = n_local * ncols
    Mstore = torch.zeros((max_basis, flat_n), dtype=acc, device=dev)
    # XB row 0 = A·x0 margins (f64); row j >= 1 = A·v_j margins of gradient
    # basis j. Holding them in ONE buffer lets the whole G row for a new
    # basis vector come from a single dgemv instead of a dgemv + two dots.
    XB = torch.zeros((max_basis + 1, flat_n), dtype=torch.float64, device=dev)
    XB[0] = xm0.to(torch.float64)
    T = 0  # gradient basis vectors so far

    def new_basis_async(m_t: torch.Tensor) -> Tuple[int, torch.Tensor, torch.Tensor]:
        """Enqueue registration of gradient basis vector v = Aᵀ m_t / c:
        returns (index, gm, row_device) with the G-row still on device —
        call finish_basis(index, row_device) before using G entries for it."""
        nonlocal T
        if T >= max_basis:
            raise RuntimeError("gram basis overflow — raise max_basis")
        m_global = op.all_gather_m(m_t, ncols)
        gm = op.matvec(m_global, ncols)
        gm = ops.axpby(1.0 / c, gm, 0.0, gm)  # gm = A·v (local slice)
        t = T + 1  # G index
        # G row: dots of v_t with x0 and all previous v_j (fp64, allreduced)
        md = m_t.to(torch.float64)
        XB[t] = gm.to(torch.float64)
        row = XB[: t + 1] @ md  # dots with x0 and every basis incl. the new one
        comm.allreduce_(row)
        Mstore[T] = m_t
        T += 1
        return t, gm, row

    def finish_basis_host(t: int, row_h: np.ndarray) -> None:
        G[t, : t + 1] = row_h
        G[: t + 1, t] = row_h

    def finish_basis(t: int, row: torch.Tensor) -> None:
        finish_basis_host(t, (row / c).cpu().numpy())

    def quad(a: np.ndarray, b: np.ndarray, k: int) -> float:
        return float(a[:k] @ (G[:k, :k] @ b[:k]))

    def eval_loss_async(vm: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """Enqueue multiplier/loss at tracked margins vm (zero data passes);
        returns (loss_count_device, multiplier)."""
        mult, lc = gradient.multiplier_loss(data, vm)
        comm.allreduce_(lc)
        return lc, mult

    def read_loss(lc: torch.Tensor) -> float:
        return float(lc[0]) / c

    def eval_loss(vm: torch.Tensor) -> Tuple[float, torch.Tensor]:
        lc, mult = eval_loss_async(vm)
        return read_loss(lc), mult

    def coef(vec_len: int) -> np.ndarray:
        return np.zeros(vec_len)

    NB = max_basis + 1
    cx, cz = coef(NB), coef(NB)
    cx[0] = 1.0
    cz[0] = 1.0
    xm = xm0.clone()
    zm = xm0.clone()
    theta = math.inf
    L = L0
    backtrack_simple = True
    loss_history: List[float] = []

    # Fused trial fast path (GPU, binary loss, affine prox, f32 margins):
    # k_multiplier_affine evaluates y without materializing ym, and
    # k_gram_state_update finishes the basis registration + both margin
    # updates in ONE pass — ~12 small launches collapse to ~5 around the
    # dominant K·m stream.
    use_fused = (
        dev.type == "cuda"
        and acc == torch.float32
        and (getattr(gradient, "LOSS_TYPE", -1) >= 0 or ncols > 1)
        and updater.PROX_KIND in (ops.PROX_SIMPLE, ops.PROX_SQUARED_L2)
        and ops._use_hip(data.features)
    )
    if use_fused:
        from .ops import hiplib as _hl

        md_buf = torch.empty(flat_n, dtype=torch.float64, device=dev)
        lc_x_buf = torch.empty(2, dtype=torch.float64, device=dev)
        sw = getattr(data, "sample_weight", None)
        labels_f32 = data.labels.to(torch.float32).contiguous()

    def fused_y_trial(th: float, L_now: float, xm_o, zm_o, cz_o, cx_o):
        """One fused backtracking trial at theta=th: returns
        (t_y, step, cz_t, cx_t, zm_new, xm_new, lc_y, row_dev)."""
        nonlocal T
        if T >= max_basis:
            raise RuntimeError("gram basis overflow — raise max_basis")
        count_eval = None
        if ncols == 1:
            m_y = torch.empty(flat_n, dtype=acc, device=dev)
            lc_y = torch.empty(2, dtype=torch.float64, device=dev)
            _hl.gram_mult_affine(xm_o, zm_o, 1.0 - th, th, labels_f32,
                                 gradient.LOSS_TYPE, sw, m_y, lc_y)
            comm.allreduce_(lc_y)
            # the fused path bypasses gradient.multiplier_loss, so notify a
            # counting wrapper (bench.py) that a loss evaluation happened
            count_eval = getattr(gradient, "count_eval", None)
            if count_eval is not None:
                count_eval()
        else:
            # multiclass: the softmax multiplier kernel needs materialized
            # margins; only the state-update fusion applies
            ym = ops.axpby(1.0 - th, xm_o, th, zm_o)
            lc_y, m_y = eval_loss_async(ym)
        m_global = op.all_gather_m(m_y, ncols)
        gm_raw = op.matvec(m_global, ncols)  # unscaled K·m; state kernel scales by 1/c
        t = T + 1
        step = 1.0 / (th * L_now)
        pz = (1.0 - step * reg_param
              if updater.PROX_KIND == ops.PROX_SQUARED_L2 else 1.0)
        zm_new = torch.empty_like(zm_o)
        xm_new = torch.empty_like(xm_o)
        _hl.gram_state_update(gm_raw, m_y, xm_o, zm_o, 1.0 / c, th, pz, -step,
                              XB[t], md_buf, Mstore[T], zm_new, xm_new)
        row = XB[: t + 1] @ md_buf
        comm.allreduce_(row)
        T += 1
        cz_t = prox_coeff(cz_o, t, step)
        cx_t = (1.0 - th) * cx_o + th * cz_t
        lc_x = None
        if backtrack_simple and beta < 1.0:
            if ncols == 1:
                # loss-only x-eval through the same kernel (a=1, b=0); m_y's
                # buffer is free again (already persisted into Mstore/md)
                _hl.gram_mult_affine(xm_new, xm_new, 1.0, 0.0, labels_f32,
                                     gradient.LOSS_TYPE, sw, m_y, lc_x_buf)
                comm.allreduce_(lc_x_buf)
                if count_eval is not None:
                    count_eval()
                lc_x = lc_x_buf
            else:
                lc_x, _ = eval_loss_async(xm_new)
        return t, step, cz_t, cx_t, zm_new, xm_new, lc_y, row, lc_x

    def prox_coeff(cz_old: np.ndarray, t_idx: int, step: float) -> np.ndarray:
        out = cz_old.copy()
        if updater.PROX_KIND == ops.PROX_SQUARED_L2:
            out *= (1.0 - step * reg_param)
        out[t_idx] += -step
        return out

    def reg_value_from_norm(csel: np.ndarray, k: int) -> float:
        if updater.PROX_KIND == ops.PROX_SQUARED_L2 and reg_param > 0:
            return 0.5 * reg_param * max(quad(csel, csel, k), 0.0)
        return 0.0

    broke = False
    for n_iter in range(1, num_iterations + 1):
        t_iter0 = time.perf_counter()
        cx_old, cz_old = cx.copy(), cz.copy()
        xm_old, zm_old = xm, zm
        L_old = L
        L = L * alpha
        theta_old = theta

        f_y = 0.0
        f_x_bt: Optional[float] = None
        cy = None
        t_y = 0
        n_backtracks = 0

        while True:
            theta = 2.0 / (1.0 + math.sqrt(1.0 + 4.0 * (L / L_old) / (theta_old * theta_old)))
            cy = (1.0 - theta) * cx_old + theta * cz_old
            lc_x = None
            if use_fused:
                (t_y, step, cz, cx, zm, xm, lc_y, row_y, lc_x) = fused_y_trial(
                    theta, L, xm_old, zm_old, cz_old, cx_old)
            else:
                ym = ops.axpby(1.0 - theta, xm_old, theta, zm_old)
                # Enqueue the whole trial's device work before the first host
                # read: y-loss, basis registration (K·m + G row), the margin
                # updates, and (simple mode) the x-loss — then pay ONE
                # pipeline wait instead of three.
                lc_y, m_y = eval_loss_async(ym)
                t_y, gm_y, row_y = new_basis_async(m_y)
                step = 1.0 / (theta * L)
                cz = prox_coeff(cz_old, t_y, step)
                cx = (1.0 - theta) * cx_old + theta * cz
                zm = updater.prox_margins(zm_old, gm_y, step, reg_param)
                xm = ops.axpby(1.0 - theta, xm_old, theta, zm)

            if beta >= 1.0:
                packed = torch.cat([lc_y, row_y]).cpu().numpy()
                f_y = packed[0] / c
                finish_basis_host(t_y, packed[2:] / c)
                break

            if backtrack_simple and lc_x is None:
                lc_x, _ = eval_loss_async(xm)
            # ONE packed D2H transfer for everything this trial must read on
            # the host (y-loss, the new G row, and the pre-enqueued x-loss)
            # instead of three small syncs.
            packed = torch.cat([lc_y, row_y] + ([lc_x] if lc_x is not None
                                                else [])).cpu().numpy()
            f_y = packed[0] / c
            finish_basis_host(t_y, packed[2: 3 + t_y] / c)

            k = T + 1
            dxy = cx - cy
            xy_sq = max(quad(dxy, dxy, k), 0.0)
            if xy_sq == 0.0:
                break

            if backtrack_simple:
                # x == y is excluded above, so reading the pre-enqueued f_x
                # here matches the reference's compute-f_x-after-the-check
                # order (and at xy_sq == 0 its value would equal f_y anyway).
                f_x = packed[3 + t_y] / c
                f_x_bt = f_x
                xy_dot_gy = float(dxy[:k] @ G[:k, t_y])
                q_x = f_y + xy_dot_gy + 0.5 * L * xy_sq
                localL = L + 2.0 * max(f_x - q_x, 0.0) / xy_sq
                backtrack_simple = abs(f_y - f_x) >= backtrack_tol * max(abs(f_x), abs(f_y))
            else:
                f_x, m_x = eval_loss(xm)
                f_x_bt = f_x
                t_x, _gm_x, row_x = new_basis_async(m_x)
                finish_basis(t_x, row_x)
                k = T + 1
                localL = 2.0 * float(dxy[:k] @ (G[:k, t_x] - G[:k, t_y])) / xy_sq

            if localL <= L or L >= Lexact:
                break
            n_backtracks += 1
            if not math.isinf(localL):
                L = min(Lexact, localL)
            else:
                localL = L
            L = min(Lexact, max(localL, L / beta))

        k = T + 1
        # loss history (reference semantics; all modes are pass-free here)
        if loss_history_mode in ("exact", "backtrack") and f_x_bt is not None:
            loss_history.append(f_x_bt + reg_value_from_norm(cx, k))
        elif loss_history_mode in ("exact", "backtrack"):
            f_x2, _ = eval_loss(xm)
            loss_history.append(f_x2 + reg_value_from_norm(cx, k))
        else:
            loss_history.append(f_y + reg_value_from_norm(cy, k))

        if math.isnan(f_y) or math.isinf(f_y):
            logger.warning("Unable to compute loss function.")
            broke = True

        dx = cx - cx_old
        norm_x = math.sqrt(max(quad(cx, cx, k), 0.0))
        norm_dx = math.sqrt(max(quad(dx, dx, k), 0.0))
        restarted = False
        if not broke:
            if norm_dx == 0.0 and n_iter > 1:
                broke = True
            elif norm_dx < convergence_tol * max(norm_x, 1.0):
                broke = True

        if not broke and may_restart and float(dx[:k] @ G[:k, t_y]) > 0.0:
            cz = cx.copy()
            zm = xm.clone()
            theta = math.inf
            backtrack_simple = True
            restarted = True

        if metrics is not None:
            metrics.log(iter=n_iter, loss=loss_history[-1], f_y=f_y, L=L,
                        theta=theta, n_backtracks=n_backtracks,
                        restarted=restarted, norm_dx=norm_dx, solver="gram",
                        basis_size=T, iter_seconds=time.perf_counter() - t_iter0)
        if broke:
            break
        if iteration_hook is not None and iteration_hook(n_iter) == "stop":
            break

    # materialize x = cx0·x0 + Aᵀ(Σ_j cx_j m_j)/c
    if T > 0:
        coefs = torch.from_numpy(cx[1: T + 1]).to(device=dev, dtype=acc)
        mcomb = coefs @ Mstore[:T]
        if ncols > 1:
            from .ops.multiclass import grad_from_mult_multi

            u = grad_from_mult_multi(feats, mcomb, gradient.num_classes)
        else:
            u = ops.dense_grad_from_mult(feats, mcomb)
        comm.allreduce_(u)
        x = (cx[0] * x0.to(acc) + u / c).to(initial_weights.dtype)
    else:
        x = (cx[0] * x0).to(initial_weights.dtype)

    logger.info("run_gram finished: %d iterations, basis size %d, K build %.2fs",
                len(loss_history), T, op.build_seconds)
    return x, loss_history
