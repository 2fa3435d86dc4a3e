"""Multinomial (softmax) logistic regression ops — oracle + dispatch.

A model family beyond the reference (MLlib 1.3's LogisticGradient is binary):
weights W [d, K] (stored flattened feature-major, classes contiguous),
margins Z [n, K] = A @ W, labels are class indices in {0..K-1}:

  p_i = softmax(z_i);  loss_i = logsumexp(z_i) - z_{i, y_i}
  M[i, k] = p_{i,k} - 1[y_i == k];  grad = Aᵀ M   (shape [d, K])

The GPU kernels work on a class dimension padded to KC = ceil(K/4)*4 (16-B
alignment of the per-feature class row); padding classes carry -inf margins
conceptually — implemented by restricting every softmax/loss loop to the
logical K, with padded multiplier columns exactly zero. Padding/unpadding
happens in this layer; callers only see the logical [d*K] / [n*K] flats.

Composes with the existing machinery: mini-batch masks, per-example sample
weights, and margin-state tracking (margins are linear in W and the prox
operators act elementwise on the flattened weights).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import _use_hip, reference as _ref  # noqa: F401  (dispatch helper)


def _pad_classes(t2d: torch.Tensor, kc: int) -> torch.Tensor:
    n, k = t2d.shape
    if k == kc:
        return t2d.contiguous()
    out = torch.zeros((n, kc), dtype=t2d.dtype, device=t2d.device)
    out[:, :k] = t2d
    return out


def padded_k(k: int) -> int:
    """Class-dim padding: snapped to the VALU kernel template sizes
    (4/8/16/32) for small K; plain ceil-4 padding above 32, where the GPU
    path runs GEMM-shaped (hipBLASLt margins/grad + torch multiplier stage —
    bf16 or f32 shards)."""
    for kc in (4, 8, 16, 32):
        if k <= kc:
            return kc
    return (k + 3) // 4 * 4


# --- oracle (plain torch; CPU tier + GPU-kernel ground truth) ---

def ref_margins_multi(features: torch.Tensor, wflat: torch.Tensor, k: int) -> torch.Tensor:
    acc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16, torch.float8_e4m3fn) else features.dtype
    w = wflat.reshape(features.shape[1], k).to(acc)
    return (features.to(acc) @ w).reshape(-1)  # flat [n*k]


def ref_eval_multi_from_margins(
    margins_flat: torch.Tensor,
    labels: torch.Tensor,
    features: torch.Tensor,
    k: int,
    mask: Optional[torch.Tensor] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[Optional[torch.Tensor], torch.Tensor]:
    n = features.shape[0]
    z = margins_flat.reshape(n, k)
    acc = z.dtype
    y = labels.to(torch.int64)
    lse = torch.logsumexp(z, dim=1)
    zy = z.gather(1, y.unsqueeze(1)).squeeze(1)
    loss = lse - zy
    m = torch.softmax(z, dim=1)
    m = m.scatter_add(1, y.unsqueeze(1), -torch.ones((n, 1), dtype=acc, device=z.device))
    scale = torch.ones(n, dtype=acc, device=z.device)
    count_t = None
    if mask is not None:
        scale = scale * mask.to(acc)
    if sample_weight is not None:
        scale = scale * sample_weight.to(acc)
    if mask is not None or sample_weight is not None:
        loss = loss * scale
        m = m * scale.unsqueeze(1)
        count_t = scale.to(torch.float64).sum()
    if count_t is None:
        count_t = torch.tensor(float(n), dtype=torch.float64, device=z.device)
    loss_count = torch.stack([loss.to(torch.float64).sum(), count_t])
    if not need_grad:
        return None, loss_count
    facc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16, torch.float8_e4m3fn) else features.dtype
    grad = (features.to(facc).T @ m.to(facc)).reshape(-1)  # flat [d*k]
    return grad, loss_count


def ref_eval_multi(features, labels, wflat, k, mask=None, need_grad=True,
                   sample_weight=None):
    zf = ref_margins_multi(features, wflat, k)
    return ref_eval_multi_from_margins(zf, labels, features, k, mask,
                                       need_grad, sample_weight)


def ref_multiplier_multi(
    z: torch.Tensor,  # [n, k] logical margins
    labels: torch.Tensor,
    mask: Optional[torch.Tensor] = None,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """(M [n,k], loss_count f64[2]) — the n-space softmax multiplier stage."""
    n, k = z.shape
    acc = z.dtype
    y = labels.to(torch.int64)
    lse = torch.logsumexp(z, dim=1)
    loss = lse - z.gather(1, y.unsqueeze(1)).squeeze(1)
    m = torch.softmax(z, dim=1)
    m = m.scatter_add(1, y.unsqueeze(1),
                      -torch.ones((n, 1), dtype=acc, device=z.device))
    count_t = None
    if mask is not None or sample_weight is not None:
        scale = torch.ones(n, dtype=acc, device=z.device)
        if mask is not None:
            scale = scale * mask.to(acc)
        if sample_weight is not None:
            scale = scale * sample_weight.to(acc)
        loss = loss * scale
        m = m * scale.unsqueeze(1)
        count_t = scale.to(torch.float64).sum()
    if count_t is None:
        count_t = torch.tensor(float(n), dtype=torch.float64, device=z.device)
    return m, torch.stack([loss.to(torch.float64).sum(), count_t])


# --- CSR shards (oracle; GPU kernels are k_csr_margins_multi/k_csc_grad_multi) ---

def _csr_rows(rowptr: torch.Tensor) -> torch.Tensor:
    n = rowptr.numel() - 1
    counts = torch.diff(rowptr.to(torch.int64))
    return torch.repeat_interleave(
        torch.arange(n, device=rowptr.device, dtype=torch.int64), counts)


def ref_csr_margins_multi(rowptr, col, val, wflat, k: int, d: int) -> torch.Tensor:
    acc = torch.float32 if val.dtype in (torch.bfloat16, torch.float16) else val.dtype
    w = wflat.reshape(d, k).to(acc)
    n = rowptr.numel() - 1
    z = torch.zeros((n, k), dtype=acc, device=val.device)
    z.index_add_(0, _csr_rows(rowptr),
                 val.to(acc).unsqueeze(1) * w[col.to(torch.int64)])
    return z.reshape(-1)


def ref_csr_grad_multi(rowptr, col, val, m2d: torch.Tensor, d: int) -> torch.Tensor:
    acc = m2d.dtype
    grad = torch.zeros((d, m2d.shape[1]), dtype=acc, device=val.device)
    grad.index_add_(0, col.to(torch.int64),
                    val.to(acc).unsqueeze(1) * m2d[_csr_rows(rowptr)])
    return grad.reshape(-1)


# --- dispatch (GPU -> HIP kernels via hiplib; CPU -> oracle) ---

def _gemm_margins_multi(features: torch.Tensor, wflat: torch.Tensor, k: int,
                        kc: int) -> torch.Tensor:
    """Margins as a skinny hipBLASLt GEMM: Z[n,KC] = A[n,d]·bf16(W)[d,KC].

    This is the idiomatic route for GEMM-shaped work — the per-element K-fma
    VALU kernel is load-issue bound at ~12x the A-stream floor for K=16
    (profiles/r01_multiclass_trace.txt). W is rounded to bf16 (the standard
    bf16 mixed-precision compute path; A is already bf16, accumulation f32);
    SPARKAGD_MULTI_MARGINS=valu selects the exact-f32-weights kernel instead.
    """
    from . import hiplib

    n, d = features.shape
    wt = torch.zeros((kc, d), dtype=torch.bfloat16, device=features.device)
    wt[:k] = wflat.reshape(d, k).T.to(torch.bfloat16)
    z = torch.empty((n, kc), dtype=torch.float32, device=features.device)
    hiplib.gemm_bf16f32_nt(features, wt, z)
    return z.reshape(-1)


def margins_multi(features: torch.Tensor, wflat: torch.Tensor, k: int) -> torch.Tensor:
    """Flat padded margins [n*KC] (opaque to callers; feed back into
    eval_multi_from_margins / axpby for margin tracking)."""
    import os

    kc = padded_k(k)
    if _use_hip(features):
        from . import hiplib

        algo = os.environ.get("SPARKAGD_MULTI_MARGINS", "auto")
        if kc > 32:
            # VALU kernels are KC-templated <= 32; large K runs GEMM-shaped
            if features.dtype == torch.bfloat16 and algo != "valu":
                return _gemm_margins_multi(features, wflat, k, kc)
            if features.dtype == torch.float32:
                # rocBLAS via torch.mm is the idiomatic f32 library GEMM
                z = features @ wflat.reshape(features.shape[1], k).to(torch.float32)
                return _pad_classes(z, kc).reshape(-1)
            raise NotImplementedError(
                f"K={k} > 32 margins need a bf16 (hipBLASLt) or f32 (rocBLAS) shard")
        if features.dtype == torch.bfloat16 and algo in ("auto", "gemm"):
            return _gemm_margins_multi(features, wflat, k, kc)
        return hiplib.dense_margins_multi(features, wflat, k, kc)
    z = ref_margins_multi(features, wflat, k).reshape(features.shape[0], k)
    return _pad_classes(z, kc).reshape(-1)


def _eval_large_k_from_margins(features, margins_padded_flat, labels, k, kc,
                               mask, need_grad, sample_weight):
    """K > 32 path: the generic-K multiplier kernel (wave-per-row
    k_multiplier_multi_anyk — ONE launch instead of the ~6-kernel torch
    stage this replaced in round 1), then grad as the hipBLASLt TN GEMM
    (bf16 shards, padded M fed straight in — its pad columns are exact
    zeros) or rocBLAS torch.mm (f32 shards)."""
    import os

    from . import hiplib

    n, d = features.shape
    M, loss_count = hiplib.multiplier_multi(margins_padded_flat, labels, k, kc,
                                            mask, sample_weight)
    if not need_grad:
        return None, loss_count
    if (features.dtype == torch.bfloat16
            and os.environ.get("SPARKAGD_MULTI_GRAD", "auto") != "valu"):
        grad = hiplib.gemm_bf16f32_tn(features, M.reshape(n, kc))
        return grad.reshape(d, kc)[:, :k].reshape(-1).contiguous(), loss_count
    if features.dtype == torch.float32:
        m = M.reshape(n, kc)[:, :k]
        return (features.T @ m).reshape(-1), loss_count
    raise NotImplementedError(
        f"K={k} > 32 gradient needs a bf16 (hipBLASLt) or f32 (rocBLAS) shard")


def eval_multi_from_margins(features, margins_padded_flat, labels, k,
                            mask=None, need_grad=True, sample_weight=None):
    kc = padded_k(k)
    if _use_hip(features):
        from . import hiplib

        if kc > 32:
            return _eval_large_k_from_margins(
                features, margins_padded_flat, labels, k, kc, mask, need_grad,
                sample_weight)
        return hiplib.dense_eval_multi_from_margins(
            features, margins_padded_flat, labels, k, kc, mask, need_grad,
            sample_weight)
    n = features.shape[0]
    z = margins_padded_flat.reshape(n, kc)[:, :k].reshape(-1)
    grad, lc = ref_eval_multi_from_margins(z, labels, features, k, mask,
                                           need_grad, sample_weight)
    return grad, lc


def eval_multi(features, labels, wflat, k, mask=None, need_grad=True,
               sample_weight=None):
    zf = margins_multi(features, wflat, k)
    return eval_multi_from_margins(features, zf, labels, k, mask, need_grad,
                                   sample_weight)


# --- CSR dispatch ---

def csr_margins_multi(shard, wflat: torch.Tensor, k: int) -> torch.Tensor:
    """Flat padded margins [n*KC] over a CSRShard. K > 32 runs the KC=32
    gather kernel per 32-class chunk (the nnz stream re-reads per chunk —
    the K-proportional W/M traffic dominates anyway)."""
    kc = padded_k(k)
    if _use_hip(shard.val):
        from . import hiplib

        if kc <= 32:
            return hiplib.csr_margins_multi(shard.rowptr, shard.col, shard.val,
                                            wflat, k, kc, shard.d)
        n, d = shard.n, shard.d
        w2 = wflat.reshape(d, k).to(torch.float32)
        z = torch.zeros((n, kc), dtype=torch.float32, device=shard.val.device)
        for lo in range(0, k, 32):
            hi = min(lo + 32, k)
            wc = w2[:, lo:hi]
            zc = hiplib.csr_margins_multi(shard.rowptr, shard.col, shard.val,
                                          wc.reshape(-1), hi - lo,
                                          padded_k(hi - lo), d)
            z[:, lo:hi] = zc.reshape(n, padded_k(hi - lo))[:, : hi - lo]
        return z.reshape(-1)
    z = ref_csr_margins_multi(shard.rowptr, shard.col, shard.val, wflat, k,
                              shard.d).reshape(-1, k)
    return _pad_classes(z, kc).reshape(-1)


def eval_multi_csr_from_margins(shard, margins_padded_flat, k,
                                mask=None, need_grad=True, sample_weight=None):
    kc = padded_k(k)
    n = shard.n
    if _use_hip(shard.val):
        from . import hiplib

        # the multiplier kernel covers every KC (templated <= 32, generic
        # wave-per-row above)
        M, lc = hiplib.multiplier_multi(margins_padded_flat, shard.labels,
                                        k, kc, mask, sample_weight)
        if not need_grad:
            return None, lc
        if shard.csc is None:
            raise RuntimeError(
                "CSR multiclass gradient needs the deterministic CSC copy "
                "(CSRShard(..., deterministic=True))")
        colptr, crow, cval = shard.csc
        if kc <= 32:
            gradp = hiplib.csc_grad_multi(colptr, crow, cval, M, shard.d, kc,
                                          getattr(shard, 'csc_heavy', None))
            if kc != k:
                return gradp.reshape(shard.d, kc)[:, :k].reshape(-1).contiguous(), lc
            return gradp, lc
        m2d = M.reshape(n, kc)[:, :k]
        grad = torch.empty((shard.d, k), dtype=torch.float32,
                           device=shard.val.device)
        for lo in range(0, k, 32):
            hi = min(lo + 32, k)
            kcc = padded_k(hi - lo)
            mc_chunk = torch.zeros((n, kcc), dtype=torch.float32,
                                   device=shard.val.device)
            mc_chunk[:, : hi - lo] = m2d[:, lo:hi]
            gc = hiplib.csc_grad_multi(colptr, crow, cval,
                                       mc_chunk.reshape(-1), shard.d, kcc,
                                       getattr(shard, 'csc_heavy', None))
            grad[:, lo:hi] = gc.reshape(shard.d, kcc)[:, : hi - lo]
        return grad.reshape(-1).contiguous(), lc
    z = margins_padded_flat.reshape(n, kc)[:, :k]
    m, lc = ref_multiplier_multi(z, shard.labels, mask, sample_weight)
    if not need_grad:
        return None, lc
    return ref_csr_grad_multi(shard.rowptr, shard.col, shard.val, m, shard.d), lc


def eval_multi_csr(shard, wflat, k, mask=None, need_grad=True,
                   sample_weight=None):
    zf = csr_margins_multi(shard, wflat, k)
    return eval_multi_csr_from_margins(shard, zf, k, mask, need_grad,
                                       sample_weight)


# --- n-space pieces for the Gram (dual-space) solver ---

def multiplier_loss_multi(labels, margins_padded_flat, k, mask=None,
                          sample_weight=None):
    """(M padded flat [n*KC] with exact-zero pad columns, loss_count f64[2])
    from padded margins — zero data passes; the multiclass analog of
    ops.dense_multiplier_loss for the Gram solver."""
    kc = padded_k(k)
    n = margins_padded_flat.numel() // kc
    if margins_padded_flat.is_cuda:
        import os

        if os.environ.get("SPARKAGD_FORCE_REFERENCE") != "1":
            from . import hiplib

            return hiplib.multiplier_multi(margins_padded_flat, labels, k, kc,
                                           mask, sample_weight)
    z = margins_padded_flat.reshape(n, kc)[:, :k]
    m, lc = ref_multiplier_multi(z, labels, mask, sample_weight)
    return _pad_classes(m, kc).reshape(-1), lc


def grad_from_mult_multi(features, m_padded_flat, k):
    """grad flat [d*k] = Aᵀ·M from the padded multiplier (dense shards)."""
    kc = padded_k(k)
    n, d = features.shape
    if _use_hip(features):
        import os

        from . import hiplib

        if (features.dtype == torch.bfloat16
                and os.environ.get("SPARKAGD_MULTI_GRAD", "auto") != "valu"):
            gradp = hiplib.gemm_bf16f32_tn(
                features, m_padded_flat.reshape(n, kc)).reshape(-1)
        else:
            if kc > 32:
                raise NotImplementedError(
                    "K > 32 VALU grad unsupported — bf16 shards use the GEMM")
            lib = hiplib.load()
            n_rb = int(lib.agd_multi_rowblocks(n, d, kc))
            gradp = torch.empty(d * kc, dtype=torch.float32,
                                device=features.device)
            part = (torch.empty(n_rb * d * kc, dtype=torch.float32,
                                device=features.device)
                    if n_rb > 1 else gradp)
            rc = lib.agd_grad_multi(hiplib._ptr(features),
                                    hiplib._DTYPE_CODE[features.dtype],
                                    hiplib._ptr(m_padded_flat.contiguous()),
                                    n, d, kc, hiplib._ptr(part), n_rb,
                                    hiplib._ptr(gradp),
                                    hiplib._stream(features))
            hiplib._check(rc)
        if kc != k:
            return gradp.reshape(d, kc)[:, :k].reshape(-1).contiguous()
        return gradp
    acc = torch.float32 if features.dtype in (torch.bfloat16, torch.float16) else features.dtype
    m = m_padded_flat.reshape(n, kc)[:, :k].to(acc)
    return (features.to(acc).T @ m).reshape(-1)
