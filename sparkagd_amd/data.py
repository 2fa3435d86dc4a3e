"""GPU-resident data shards + seeded synthetic generators.

The MI355X-native replacement for the reference's data plane: where the
reference holds an ``RDD[(Double, Vector)]`` of cached partitions and ships
the weight vector by torrent broadcast every evaluation
(``AcceleratedGradientDescent.scala:128,193``; ``Suite.scala:51``), here each
rank (one process per GPU) holds one shard pinned in its 288 GB of HBM3E and
the weight vector is device-resident from the start — there is no per-
iteration host->device weight traffic at all (the analog of the reference's
"task closure < 1 MB" cluster test, ``Suite.scala:244-259``).

Row sharding only (features are never split across GPUs): the per-iteration
collective is then a single all-reduce of (grad_sum ‖ loss ‖ count).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

from . import ops


class DenseShard:
    """One rank's rows of a dense design matrix, resident on its device.

    features: [n_local, d] (bf16 / f32 / f64), row-major contiguous.
    labels:   [n_local] float32 (float64 on CPU shards is also accepted).
    """

    kind = "dense"

    def __init__(self, features: torch.Tensor, labels: torch.Tensor,
                 sample_weight: Optional[torch.Tensor] = None):
        if features.ndim != 2:
            raise ValueError("features must be [n, d]")
        if labels.ndim != 1 or labels.shape[0] != features.shape[0]:
            raise ValueError("labels must be [n]")
        if not features.is_contiguous():
            features = features.contiguous()
        self.features = features
        self.labels = labels.to(device=features.device)
        if self.labels.dtype not in (torch.float32, torch.float64):
            self.labels = self.labels.to(torch.float32)
        # optional per-example weights (weighted GLM; count becomes sum of
        # weights so all mean-loss/mean-gradient semantics carry over)
        self.sample_weight = None
        if sample_weight is not None:
            if sample_weight.shape[0] != features.shape[0]:
                raise ValueError("sample_weight must be [n]")
            self.sample_weight = sample_weight.to(
                device=features.device, dtype=torch.float32).contiguous()

    @property
    def n(self) -> int:
        return self.features.shape[0]

    @property
    def d(self) -> int:
        return self.features.shape[1]

    @property
    def device(self) -> torch.device:
        return self.features.device

    @property
    def nbytes(self) -> int:
        return self.features.numel() * self.features.element_size() + self.labels.numel() * self.labels.element_size()

    def eval(self, w: torch.Tensor, loss_type: int, mask: Optional[torch.Tensor] = None,
             need_grad: bool = True):
        return ops.dense_eval(self.features, self.labels, w, loss_type, mask,
                              need_grad, self.sample_weight)

    # --- margin-state tracking support (one data pass instead of two when
    # the caller already holds A @ w; see optimizer.py 'track_margins') ---
    def margins(self, v: torch.Tensor) -> torch.Tensor:
        return ops.dense_margins(self.features, v)

    def eval_from_margins(self, margins: torch.Tensor, loss_type: int,
                          mask: Optional[torch.Tensor] = None, need_grad: bool = True):
        return ops.dense_eval_from_margins(self.features, margins, self.labels,
                                           loss_type, mask, need_grad,
                                           self.sample_weight)


class CSRShard:
    """One rank's rows of a CSR sparse design matrix.

    rowptr: [n_local+1] int32/int64, col: [nnz] int32, val: [nnz] f32,
    labels: [n_local] f32. ``d`` is the (global) feature dimension.
    """

    kind = "csr"

    def __init__(self, rowptr: torch.Tensor, col: torch.Tensor, val: torch.Tensor,
                 labels: torch.Tensor, d: int, deterministic: bool = True,
                 sample_weight: Optional[torch.Tensor] = None):
        # canonical int32 indices (the HIP kernels' index width); guard the
        # ranges instead of silently truncating
        if val.numel() > 2**31 - 1 or d > 2**31 - 1:
            raise ValueError("CSR shard exceeds int32 index range — split it")
        self.rowptr = rowptr.contiguous().to(torch.int32)
        self.col = col.contiguous().to(torch.int32)
        self.val = val.contiguous()
        self.labels = labels.to(device=val.device)
        if self.labels.dtype not in (torch.float32, torch.float64):
            self.labels = self.labels.to(torch.float32)
        self._d = int(d)
        self.sample_weight = None
        if sample_weight is not None:
            self.sample_weight = sample_weight.to(
                device=val.device, dtype=torch.float32).contiguous()
        # deterministic=True builds a CSC copy of the shard at construction
        # (2x nnz memory) so the A^T·m pass is a gather instead of an fp32
        # atomic scatter: bitwise-reproducible gradients (SURVEY.md §5,
        # 'Race detection'). deterministic=False keeps the atomic path.
        self.csc = self._build_csc() if deterministic else None
        # Power-law column skew: thread-per-column serializes on hot columns
        # (a Zipf(1.1) d=1e7 shard measured 943 ms/step vs 2.26 uniform);
        # columns above CSC_HEAVY_T nnz get split into CSC_TASK_S-entry
        # wave-tasks with a deterministic in-order combine. None when no
        # column exceeds the threshold (uniform data: zero overhead).
        self.csc_heavy = self._build_csc_heavy() if self.csc is not None else None

    @property
    def n(self) -> int:
        return self.rowptr.numel() - 1

    @property
    def d(self) -> int:
        return self._d

    @property
    def nnz(self) -> int:
        return self.val.numel()

    @property
    def device(self) -> torch.device:
        return self.val.device

    @property
    def nbytes(self) -> int:
        return (self.rowptr.numel() * self.rowptr.element_size()
                + self.col.numel() * self.col.element_size()
                + self.val.numel() * self.val.element_size()
                + self.labels.numel() * self.labels.element_size())

    def _build_csc(self):
        """Column-sorted (CSC) copy: colptr [d+1] i32, row [nnz] i32, val [nnz] f32."""
        n = self.rowptr.numel() - 1
        counts = torch.diff(self.rowptr.to(torch.int64))
        rows = torch.repeat_interleave(
            torch.arange(n, device=self.col.device, dtype=torch.int32), counts
        )
        col64 = self.col.to(torch.int64)
        order = torch.argsort(col64, stable=True)
        sorted_col = col64[order]
        csc_row = rows[order].contiguous()
        csc_val = self.val[order].contiguous()
        # colptr straight from the sorted columns (bincount over 64M nnz
        # measured 90 ms on GPU; searchsorted on the already-sorted array
        # is a fraction of that)
        colptr = torch.searchsorted(
            sorted_col,
            torch.arange(self._d + 1, device=sorted_col.device,
                         dtype=torch.int64))
        return colptr.to(torch.int32).contiguous(), csc_row, csc_val

    #: heavy-column split parameters (see csc_heavy above). The light
    #: kernel's wave executes the MAX of its 64 threads' column lengths, so
    #: a lower threshold moves imbalance into the wave-parallel heavy path;
    #: env overrides allow threshold A/Bs (profiles/r02_csr_skew_ab.txt).
    CSC_HEAVY_T = int(__import__("os").environ.get("SPARKAGD_CSC_HEAVY_T", "16"))
    CSC_TASK_S = int(__import__("os").environ.get("SPARKAGD_CSC_TASK_S", "128"))

    def _build_csc_heavy(self):
        colptr = self.csc[0].to(torch.int64)
        counts = torch.diff(colptr)
        dev = self.val.device
        # NOTE a length-sorted visit order for the light kernel (to remove
        # intra-wave imbalance) was measured and REJECTED: adjacent threads
        # then read SCATTERED csc windows, losing the contiguous-window
        # coalescing that dominates (uniform 2.26 -> 3.99 ms/step;
        # profiles/r02_csr_skew_ab.txt). order stays None (identity).
        order = None
        heavy = counts > self.CSC_HEAVY_T
        if not bool(heavy.any()):
            empty = torch.zeros(0, dtype=torch.int32, device=dev)
            return {
                "heavy_T": self.CSC_HEAVY_T, "S": self.CSC_TASK_S,
                "cols": empty,
                "taskptr": torch.zeros(1, dtype=torch.int32, device=dev),
                "task_idx": empty, "partial": None, "order": order,
            }
        cols = torch.nonzero(heavy, as_tuple=False).reshape(-1)
        ntasks = (counts[cols] + self.CSC_TASK_S - 1) // self.CSC_TASK_S
        taskptr = torch.zeros(cols.numel() + 1, dtype=torch.int64,
                              device=cols.device)
        torch.cumsum(ntasks, 0, out=taskptr[1:])
        task_idx = torch.repeat_interleave(
            torch.arange(cols.numel(), device=cols.device), ntasks)
        return {
            "heavy_T": self.CSC_HEAVY_T,
            "S": self.CSC_TASK_S,
            "cols": cols.to(torch.int32).contiguous(),
            "taskptr": taskptr.to(torch.int32).contiguous(),
            "task_idx": task_idx.to(torch.int32).contiguous(),
            "partial": torch.empty(int(taskptr[-1]), dtype=torch.float32,
                                   device=dev),
            "order": order,
        }

    def eval(self, w: torch.Tensor, loss_type: int, mask: Optional[torch.Tensor] = None,
             need_grad: bool = True):
        return ops.csr_eval(self.rowptr, self.col, self.val, self.labels, w,
                            loss_type, mask, self._d, csc=self.csc,
                            need_grad=need_grad,
                            sample_weight=self.sample_weight,
                            csc_heavy=self.csc_heavy)

    def margins(self, v: torch.Tensor) -> torch.Tensor:
        return ops.csr_margins(self.rowptr, self.col, self.val, v)

    def eval_from_margins(self, margins: torch.Tensor, loss_type: int,
                          mask: Optional[torch.Tensor] = None, need_grad: bool = True):
        return ops.csr_eval_from_margins(self.rowptr, self.col, self.val, margins,
                                         self.labels, loss_type, mask, self._d,
                                         csc=self.csc, need_grad=need_grad,
                                         sample_weight=self.sample_weight,
                                         csc_heavy=self.csc_heavy)


def add_intercept(shard: DenseShard) -> DenseShard:
    """Return a new DenseShard with an all-ones intercept column prepended
    (the reference suite's manual pattern, ``Suite.scala:47-49``). Copies the
    features (n x (d+1))."""
    ones = torch.ones((shard.n, 1), dtype=shard.features.dtype,
                      device=shard.features.device)
    return DenseShard(torch.cat([ones, shard.features], dim=1).contiguous(),
                      shard.labels)


# ---------------------------------------------------------------------------
# Seeded synthetic generators
# ---------------------------------------------------------------------------

def shard_range(n_global: int, rank: int, world_size: int) -> Tuple[int, int]:
    """Row range [lo, hi) owned by ``rank`` under balanced row sharding."""
    base = n_global // world_size
    rem = n_global % world_size
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    return lo, hi


def generate_logistic_data(
    a: float,
    b: float,
    n: int,
    seed: int,
    device: str | torch.device = "cpu",
    dtype: torch.dtype = torch.float64,
    intercept: bool = True,
) -> DenseShard:
    """Seeded 1-feature logistic data matching the distributional semantics of
    MLlib's ``GradientDescentSuite.generateGDInput(A, B, nPoints, seed)``
    (used by the reference suite at ``Suite.scala:46-49``): x1 ~ N(0,1),
    y = 1 if A + B*x1 + logistic_noise > 0 else 0, with an all-ones intercept
    column prepended when ``intercept``.
    """
    gen = torch.Generator(device="cpu").manual_seed(seed)
    x1 = torch.randn(n, generator=gen, dtype=torch.float64)
    u = torch.rand(n, generator=gen, dtype=torch.float64)
    noise = torch.log(u) - torch.log1p(-u)  # standard logistic
    y = ((a + b * x1 + noise) > 0).to(torch.float64)
    if intercept:
        feats = torch.stack([torch.ones_like(x1), x1], dim=1)
    else:
        feats = x1.reshape(-1, 1)
    dev = torch.device(device)
    return DenseShard(feats.to(device=dev, dtype=dtype), y.to(device=dev))


def generate_dense_problem(
    n: int,
    d: int,
    seed: int,
    loss_type: int = ops.LOSS_LOGISTIC,
    device: str | torch.device = "cpu",
    dtype: torch.dtype = torch.float32,
    chunk_rows: Optional[int] = None,
    label_noise: float = 0.1,
) -> Tuple[DenseShard, torch.Tensor]:
    """Random-init dense problem with planted weights, generated in row chunks
    so huge shards (hundreds of GB) never allocate an oversized temporary.

    Returns (shard, w_true). Labels follow the planted model:
    logistic/hinge: y = 1[z + noise > 0]; least squares: y = z + noise,
    with z = X @ w_true and w_true ~ N(0, 1/sqrt(d)).
    """
    dev = torch.device(device)
    if chunk_rows is None:
        # bound the fp32 staging buffer to ~1 GB regardless of d
        chunk_rows = max(64, min(65536, (1 << 28) // max(d, 1)))
    gen = torch.Generator(device=dev).manual_seed(seed)
    w_true = torch.randn(d, generator=gen, device=dev, dtype=torch.float32) / math.sqrt(d)
    feats = torch.empty((n, d), device=dev, dtype=dtype)
    labels = torch.empty(n, device=dev, dtype=torch.float32)
    for lo in range(0, n, chunk_rows):
        hi = min(lo + chunk_rows, n)
        blk = torch.randn((hi - lo, d), generator=gen, device=dev, dtype=torch.float32)
        z = blk @ w_true
        noise = torch.randn(hi - lo, generator=gen, device=dev, dtype=torch.float32) * label_noise
        if loss_type == ops.LOSS_LEAST_SQUARES:
            labels[lo:hi] = z + noise
        else:
            labels[lo:hi] = (z + noise > 0).to(torch.float32)
        feats[lo:hi] = blk.to(dtype)
        del blk
    return DenseShard(feats, labels), w_true


def generate_multiclass_problem(
    n: int,
    d: int,
    num_classes: int,
    seed: int,
    device: str | torch.device = "cpu",
    dtype: torch.dtype = torch.float32,
    chunk_rows: Optional[int] = None,
    label_noise: float = 0.5,
) -> Tuple[DenseShard, torch.Tensor]:
    """Planted multinomial problem: labels = argmax(X @ W* + noise);
    returns (shard, w_true_flat [d*K])."""
    dev = torch.device(device)
    if chunk_rows is None:
        chunk_rows = max(64, min(65536, (1 << 28) // max(d, 1)))
    gen = torch.Generator(device=dev).manual_seed(seed)
    w_true = torch.randn((d, num_classes), generator=gen, device=dev,
                         dtype=torch.float32) / math.sqrt(d)
    feats = torch.empty((n, d), device=dev, dtype=dtype)
    labels = torch.empty(n, device=dev, dtype=torch.float32)
    for lo in range(0, n, chunk_rows):
        hi = min(lo + chunk_rows, n)
        blk = torch.randn((hi - lo, d), generator=gen, device=dev, dtype=torch.float32)
        z = blk @ w_true
        z = z + torch.randn(z.shape, generator=gen, device=dev,
                            dtype=torch.float32) * label_noise
        labels[lo:hi] = z.argmax(dim=1).to(torch.float32)
        feats[lo:hi] = blk.to(dtype)
        del blk
    return DenseShard(feats, labels), w_true.reshape(-1)


def generate_multiclass_csr_problem(
    n: int,
    d: int,
    nnz_per_row: int,
    num_classes: int,
    seed: int,
    device: str | torch.device = "cpu",
    label_noise: float = 0.5,
) -> Tuple[CSRShard, torch.Tensor]:
    """Planted multinomial CSR problem: labels = argmax(A @ W* + noise);
    returns (shard, w_true_flat [d*K])."""
    dev = torch.device(device)
    gen = torch.Generator(device=dev).manual_seed(seed)
    nnz = n * nnz_per_row
    col = torch.randint(0, d, (nnz,), generator=gen, device=dev, dtype=torch.int32)
    col = col.view(n, nnz_per_row).sort(dim=1).values.reshape(-1).contiguous()
    val = torch.randn(nnz, generator=gen, device=dev, dtype=torch.float32)
    rowptr = torch.arange(0, nnz + 1, nnz_per_row, device=dev, dtype=torch.int32)
    w_true = torch.randn((d, num_classes), generator=gen, device=dev,
                         dtype=torch.float32) / math.sqrt(nnz_per_row)
    labels = torch.empty(n, device=dev, dtype=torch.float32)
    chunk = max(1, (1 << 22) // max(nnz_per_row * num_classes, 1))
    for lo in range(0, n, chunk):
        hi = min(lo + chunk, n)
        c = col[lo * nnz_per_row: hi * nnz_per_row].view(hi - lo, nnz_per_row).to(torch.int64)
        v = val[lo * nnz_per_row: hi * nnz_per_row].view(hi - lo, nnz_per_row)
        z = (v.unsqueeze(2) * w_true[c]).sum(dim=1)  # [rows, K]
        z = z + torch.randn(z.shape, generator=gen, device=dev,
                            dtype=torch.float32) * label_noise
        labels[lo:hi] = z.argmax(dim=1).to(torch.float32)
    return CSRShard(rowptr, col, val, labels, d), w_true.reshape(-1)


def generate_csr_problem(
    n: int,
    d: int,
    nnz_per_row: int,
    seed: int,
    loss_type: int = ops.LOSS_LOGISTIC,
    device: str | torch.device = "cpu",
    col_dist: str = "uniform",
    zipf_a: float = 1.1,
) -> Tuple[CSRShard, torch.Tensor]:
    """Random CSR problem: each row has ``nnz_per_row`` column indices drawn
    uniformly or Zipf-distributed (``col_dist="zipf"``: column popularity
    ~ rank^-a with hot columns scattered over the ID space by a seeded
    permutation — the realistic power-law regime that column-frequency
    clustering, :func:`reindex_columns`, targets), with N(0,1) values;
    planted labels as in :func:`generate_dense_problem`."""
    dev = torch.device(device)
    gen = torch.Generator(device=dev).manual_seed(seed)
    nnz = n * nnz_per_row
    if col_dist == "zipf":
        ranks = torch.arange(1, d + 1, device=dev, dtype=torch.float64)
        cdf = ranks.pow_(-float(zipf_a)).cumsum_(0)
        cdf /= cdf[-1].clone()
        u = torch.rand(nnz, generator=gen, device=dev, dtype=torch.float64)
        rank_idx = torch.searchsorted(cdf, u).clamp_(max=d - 1)
        scatter = torch.randperm(d, generator=gen, device=dev)
        col = scatter[rank_idx].to(torch.int32)
        del ranks, cdf, u, rank_idx, scatter
    elif col_dist == "uniform":
        col = torch.randint(0, d, (nnz,), generator=gen, device=dev, dtype=torch.int32)
    else:
        raise ValueError("col_dist must be uniform|zipf")
    col = col.view(n, nnz_per_row).sort(dim=1).values.reshape(-1).contiguous()
    val = torch.randn(nnz, generator=gen, device=dev, dtype=torch.float32)
    rowptr = torch.arange(0, nnz + 1, nnz_per_row, device=dev, dtype=torch.int32)
    w_true = torch.randn(d, generator=gen, device=dev, dtype=torch.float32) / math.sqrt(nnz_per_row)
    # z = A @ w_true via gather (chunked over rows)
    z = torch.empty(n, device=dev, dtype=torch.float32)
    chunk = max(1, (1 << 22) // max(nnz_per_row, 1))
    for lo in range(0, n, chunk):
        hi = min(lo + chunk, n)
        c = col[lo * nnz_per_row: hi * nnz_per_row].view(hi - lo, nnz_per_row).to(torch.int64)
        v = val[lo * nnz_per_row: hi * nnz_per_row].view(hi - lo, nnz_per_row)
        z[lo:hi] = (v * w_true[c]).sum(dim=1)
    noise = torch.randn(n, generator=gen, device=dev, dtype=torch.float32) * 0.1
    if loss_type == ops.LOSS_LEAST_SQUARES:
        labels = z + noise
    else:
        labels = (z + noise > 0).to(torch.float32)
    return CSRShard(rowptr, col, val, labels, d), w_true


class MixedShard:
    """Heterogeneous shard: dense AND CSR-sparse example blocks in one
    logical row space.

    MLlib's ``Gradient.compute`` accepts dense or sparse vectors per example
    within one RDD (invoked at ``AGD.scala:198``); the MI355X-native analog
    groups examples by representation — a preprocessing pass sorts rows into
    one dense block and one sparse block per rank (order does not matter for
    full-batch sums) — and evaluates each block with its own fused kernels,
    summing the (grad, loss, count) partials on device. Margins concatenate
    in block order, so margin-state tracking and mini-batch masks compose
    exactly as for homogeneous shards.

    Binary losses only (the multiclass gradient dispatches on homogeneous
    shard kinds). All parts must share the feature dimension and device.
    """

    kind = "mixed"

    def __init__(self, parts):
        parts = list(parts)
        if not parts:
            raise ValueError("MixedShard needs at least one part")
        d0, dev0 = parts[0].d, parts[0].device
        for p in parts:
            if p.d != d0:
                raise ValueError(f"feature-dim mismatch: {p.d} != {d0}")
            if p.device != dev0:
                raise ValueError("all parts must live on one device")
        self.parts = parts
        self._offsets = [0]
        for p in parts:
            self._offsets.append(self._offsets[-1] + p.n)
        self.labels = torch.cat([p.labels.to(parts[0].labels.dtype)
                                 for p in parts])
        sw = [getattr(p, "sample_weight", None) for p in parts]
        self.sample_weight = None
        if any(w is not None for w in sw):
            self.sample_weight = torch.cat([
                w if w is not None
                else torch.ones(p.n, dtype=torch.float32, device=dev0)
                for w, p in zip(sw, parts)])

    @property
    def n(self) -> int:
        return self._offsets[-1]

    @property
    def d(self) -> int:
        return self.parts[0].d

    @property
    def device(self) -> torch.device:
        return self.parts[0].device

    @property
    def nbytes(self) -> int:
        return sum(p.nbytes for p in self.parts)

    def _split(self, t: Optional[torch.Tensor]):
        if t is None:
            return [None] * len(self.parts)
        return [t[lo:hi].contiguous() for lo, hi in
                zip(self._offsets[:-1], self._offsets[1:])]

    def eval(self, w: torch.Tensor, loss_type: int,
             mask: Optional[torch.Tensor] = None, need_grad: bool = True):
        grad = None
        loss_count = None
        for p, m in zip(self.parts, self._split(mask)):
            g, lc = p.eval(w, loss_type, m, need_grad)
            loss_count = lc if loss_count is None else loss_count + lc
            if need_grad:
                grad = g if grad is None else grad + g.to(grad.dtype)
        return grad, loss_count

    def margins(self, v: torch.Tensor) -> torch.Tensor:
        ms = [p.margins(v) for p in self.parts]
        dt = ms[0].dtype
        for m in ms[1:]:
            dt = torch.promote_types(dt, m.dtype)
        return torch.cat([m.to(dt) for m in ms])

    def eval_from_margins(self, margins: torch.Tensor, loss_type: int,
                          mask: Optional[torch.Tensor] = None,
                          need_grad: bool = True):
        grad = None
        loss_count = None
        for p, vm, m in zip(self.parts, self._split(margins),
                            self._split(mask)):
            g, lc = p.eval_from_margins(vm, loss_type, m, need_grad)
            loss_count = lc if loss_count is None else loss_count + lc
            if need_grad:
                grad = g if grad is None else grad + g.to(grad.dtype)
        return grad, loss_count


def reindex_columns(shard: CSRShard):
    """Column-frequency clustering preprocessing (round-2 backlog item):
    renumber feature columns by descending occurrence count so the hottest
    columns occupy the lowest IDs — their weight entries then pack into a
    handful of cache lines that stay LLC/L2-resident, and a wave gathering a
    row's w[col] entries coalesces several lanes onto the same 64 B line
    (the CSR margins pass is gather-line-service-bound at d=1e7,
    profiles/r01_csr_kernel_trace.txt).

    Returns ``(clustered_shard, perm)`` where ``perm[new_id] = old_id``:
    train in the permuted space, then map weights back with
    ``unpermute_weights(w, perm)``. Power-law column distributions benefit;
    for uniformly random columns the permutation is a measured no-op by
    symmetry (every renumbering is distribution-identical — see BACKLOG).
    Deterministic (stable sorts throughout).
    """
    counts = torch.bincount(shard.col.to(torch.int64), minlength=shard.d)
    # perm[new] = old, hottest first; stable for reproducibility
    perm = torch.argsort(counts, descending=True, stable=True)
    inv = torch.empty_like(perm)
    inv[perm] = torch.arange(shard.d, device=perm.device, dtype=perm.dtype)
    new_col = inv[shard.col.to(torch.int64)]
    # re-sort columns within each row (keeps row-local gathers line-adjacent)
    n = shard.n
    counts_row = torch.diff(shard.rowptr.to(torch.int64))
    rows = torch.repeat_interleave(
        torch.arange(n, device=new_col.device, dtype=torch.int64), counts_row)
    key = rows * shard.d + new_col
    order = torch.argsort(key, stable=True)
    out = CSRShard(shard.rowptr, new_col[order].to(torch.int32),
                   shard.val[order], shard.labels, shard.d,
                   deterministic=shard.csc is not None,
                   sample_weight=shard.sample_weight)
    return out, perm


def unpermute_weights(w: torch.Tensor, perm: torch.Tensor) -> torch.Tensor:
    """Map weights trained on a column-clustered shard back to the original
    column IDs: out[perm[new]] = w[new]."""
    out = torch.empty_like(w)
    out[perm] = w
    return out


def permute_weights(w: torch.Tensor, perm: torch.Tensor) -> torch.Tensor:
    """Original-space weights -> clustered space (e.g. warm starts)."""
    return w[perm].contiguous()
