"""ctypes binding to the CDNA4 kernel library (csrc/libagd_hip.so).

The .so is a plain C-ABI HIP library built directly by hipcc for gfx950 —
no hipify, no torch C++ headers, no pybind — loaded with ctypes and called
with raw device pointers plus torch's current HIP stream. The build lives
in-tree (``sparkagd_amd/csrc/libagd_hip.so``) so it travels with the repo
snapshot; ``__graft_entry__.build()`` produces it.

Determinism notes:
* the dense A^T·m path writes private partial slabs per (row-block,
  column-slab) workgroup and reduces them in a fixed order => bitwise
  reproducible gradients across runs (and therefore across ranks given
  identical inputs).
* the CSR A^T·m path is deterministic when the shard carries its CSC copy
  (the default: a gather kernel, no atomics). Without it an fp32 atomic
  scatter is used, whose summation order is non-deterministic at the
  rounding level; tests cross-check the two (SURVEY.md §5 'Race detection').
"""

from __future__ import annotations

import ctypes
import os
from typing import Optional, Tuple

import torch

_DTYPE_CODE = {torch.bfloat16: 0, torch.float32: 1, torch.float64: 2, torch.float8_e4m3fn: 3}
_ACC_DTYPE = {torch.bfloat16: torch.float32, torch.float32: torch.float32, torch.float64: torch.float64, torch.float8_e4m3fn: torch.float32}

_lib: Optional[ctypes.CDLL] = None

SO_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", "csrc", "libagd_hip.so")


def so_path() -> str:
    return os.path.abspath(SO_PATH)


def load() -> ctypes.CDLL:
    global _lib
    if _lib is not None:
        return _lib
    path = so_path()
    if not os.path.exists(path):
        raise FileNotFoundError(f"HIP kernel library not found at {path}")
    lib = ctypes.CDLL(path)

    lib.agd_last_error.restype = ctypes.c_char_p
    lib.agd_version.restype = ctypes.c_int
    lib.agd_dense_rowblocks.restype = ctypes.c_longlong
    lib.agd_dense_rowblocks.argtypes = [ctypes.c_longlong, ctypes.c_longlong, ctypes.c_int]
    lib.agd_margin_slabs.restype = ctypes.c_int
    lib.agd_margin_slabs.argtypes = [ctypes.c_longlong, ctypes.c_longlong, ctypes.c_int, ctypes.c_int]

    P, LL, I, D = ctypes.c_void_p, ctypes.c_longlong, ctypes.c_int, ctypes.c_double
    lib.agd_dense_eval.restype = I
    lib.agd_dense_eval.argtypes = [P, I, P, P, P, P, LL, LL, P, P, P, P, P, LL, I, I, I, I, I, I, P, P]
    lib.agd_csr_eval.restype = I
    lib.agd_csr_eval.argtypes = [P, P, P, P, P, P, P, LL, LL, LL, P, P, P, P, I, P, P, P, I, I, P, P]
    lib.agd_axpby.restype = I
    lib.agd_axpby.argtypes = [D, P, D, P, P, LL, I, P]
    lib.agd_prox.restype = I
    lib.agd_prox.argtypes = [I, P, P, D, D, D, P, P, LL, I, P, P]
    lib.agd_fused_scalars.restype = I
    lib.agd_fused_scalars.argtypes = [P, P, P, P, P, LL, I, P, P]
    lib.agd_dot_diff.restype = I
    lib.agd_dot_diff.argtypes = [P, P, P, P, P, LL, I, P, P]
    lib.agd_gemm_bf16f32_nt.restype = I
    lib.agd_gemm_bf16f32_nt.argtypes = [P, P, P, LL, LL, LL, ctypes.c_float, P]
    lib.agd_gemm_bf16f32_tn.restype = I
    lib.agd_gemm_bf16f32_tn.argtypes = [P, P, P, LL, LL, LL, P]
    lib.agd_multi_rowblocks.restype = LL
    lib.agd_multi_rowblocks.argtypes = [LL, LL, I]
    lib.agd_margins_multi.restype = I
    lib.agd_margins_multi.argtypes = [P, I, P, LL, LL, I, P, P]
    lib.agd_multiplier_multi.restype = I
    lib.agd_multiplier_multi.argtypes = [P, P, P, P, LL, I, I, P, P, P, P]
    lib.agd_grad_multi.restype = I
    lib.agd_grad_multi.argtypes = [P, I, P, LL, LL, I, P, LL, P, P]
    lib.agd_csr_margins_multi.restype = I
    lib.agd_csr_margins_multi.argtypes = [P, P, P, P, LL, I, I, P, P]
    lib.agd_csc_grad_multi.restype = I
    lib.agd_csc_grad_multi.argtypes = [P, P, P, P, LL, I, P,
                                       I, P, P, P, LL, LL, I, P, P, P]
    lib.agd_csc_grad_skew.restype = I
    lib.agd_csc_grad_skew.argtypes = [P, P, P, P, LL, I, P, P, P, LL, LL, I, P, P, P, P]
    lib.agd_gram_mult_affine.restype = I
    lib.agd_gram_mult_affine.argtypes = [P, P, D, D, P, P, I, LL, P, P, P, P]
    lib.agd_gram_state_update.restype = I
    lib.agd_gram_state_update.argtypes = [P, P, P, P, D, D, D, D, LL, P, P, P, P, P, P]
    lib.agd_at_margin_update.restype = I
    lib.agd_at_margin_update.argtypes = [P, P, P, D, D, D, LL, I, P, P, P]

    _lib = lib
    return lib


def _check(rc: int) -> None:
    if rc != 0:
        raise RuntimeError(f"agd HIP kernel error rc={rc}: {_lib.agd_last_error().decode()}")


def _stream(t: torch.Tensor) -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream(t.device).cuda_stream)


def _ptr(t: Optional[torch.Tensor]) -> Optional[ctypes.c_void_p]:
    if t is None:
        return None
    return ctypes.c_void_p(t.data_ptr())


_red_ws_cache = {}


def _red_ws(device) -> torch.Tensor:
    """Per-device fp64 partials buffer for the two-stage scalar reductions
    (max grid 8192 x up to 5 accumulators)."""
    t = _red_ws_cache.get(device)
    if t is None:
        t = torch.empty(8192 * 5, dtype=torch.float64, device=device)
        _red_ws_cache[device] = t
    return t


def _prep_mask(mask: Optional[torch.Tensor], device) -> Optional[torch.Tensor]:
    if mask is None:
        return None
    if mask.dtype != torch.uint8:
        mask = mask.to(torch.uint8)
    return mask.contiguous().to(device)


def _prep_weights(sw: Optional[torch.Tensor], device) -> Optional[torch.Tensor]:
    if sw is None:
        return None
    if sw.dtype != torch.float32:
        sw = sw.to(torch.float32)
    return sw.contiguous().to(device)


def dense_eval(
    features: torch.Tensor,
    labels: torch.Tensor,
    w: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    lib = load()
    assert features.is_cuda and features.is_contiguous() and features.ndim == 2
    n, d = features.shape
    a_dtype = _DTYPE_CODE[features.dtype]
    acc = _ACC_DTYPE[features.dtype]
    if w.dtype != acc:
        raise TypeError(f"weights dtype {w.dtype} must be {acc} for {features.dtype} features")
    w = w.contiguous()
    labels = labels.contiguous()
    if labels.dtype != torch.float32:
        labels = labels.to(torch.float32)
    mask = _prep_mask(mask, features.device)

    dev = features.device
    loss_count = torch.zeros(2, dtype=torch.float64, device=dev)
    # margins algorithm: 0 auto, 1 VALU row-group, 2 MFMA (bf16, d%16==0)
    margins_algo = int(os.environ.get("SPARKAGD_MARGINS_ALGO", "0"))
    n_slabs = int(lib.agd_margin_slabs(n, d, a_dtype, margins_algo))
    margins = torch.empty(n_slabs * n, dtype=acc, device=dev)
    mult = torch.empty(n, dtype=acc, device=dev)
    if need_grad:
        grad = torch.empty(d, dtype=acc, device=dev)
        n_rb = int(lib.agd_dense_rowblocks(n, d, a_dtype))
        part = torch.empty(n_rb * d, dtype=acc, device=dev) if n_rb > 1 else grad
    else:
        grad = part = None
        n_rb = 1

    sw = _prep_weights(sample_weight, features.device)
    rc = lib.agd_dense_eval(
        _ptr(features), a_dtype, _ptr(labels), _ptr(mask), _ptr(sw), _ptr(w),
        n, d, _ptr(grad), _ptr(loss_count), _ptr(margins), _ptr(mult),
        _ptr(part), n_rb, loss_type, n_slabs, 1 if need_grad else 0,
        margins_algo, int(os.environ.get("SPARKAGD_NT_LOADS", "1")),  # nt A-stream: +6-9% measured
        0, _ptr(_red_ws(features.device)), _stream(features),
    )
    _check(rc)
    return grad, loss_count


def dense_margins(features: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """margins = A @ v for an arbitrary d-vector (margin-state tracking)."""
    lib = load()
    assert features.is_cuda and features.is_contiguous() and features.ndim == 2
    n, d = features.shape
    a_dtype = _DTYPE_CODE[features.dtype]
    acc = _ACC_DTYPE[features.dtype]
    if v.dtype != acc:
        raise TypeError(f"vector dtype {v.dtype} must be {acc}")
    margins_algo = int(os.environ.get("SPARKAGD_MARGINS_ALGO", "0"))
    n_slabs = int(lib.agd_margin_slabs(n, d, a_dtype, margins_algo))
    margins = torch.empty(n_slabs * n, dtype=acc, device=features.device)
    rc = lib.agd_dense_eval(
        _ptr(features), a_dtype, None, None, None, _ptr(v.contiguous()),
        n, d, None, None, _ptr(margins), None, None, 1, 0, n_slabs, 0,
        margins_algo, int(os.environ.get("SPARKAGD_NT_LOADS", "1")),
        1, _ptr(_red_ws(features.device)), _stream(features),
    )
    _check(rc)
    return margins.narrow(0, 0, n)


def dense_eval_from_margins(
    features: torch.Tensor,
    margins: torch.Tensor,
    labels: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[Optional[torch.Tensor], torch.Tensor]:
    """multiplier/loss (+ A^T·m when need_grad) from precomputed margins."""
    lib = load()
    assert features.is_cuda and features.is_contiguous() and features.ndim == 2
    n, d = features.shape
    a_dtype = _DTYPE_CODE[features.dtype]
    acc = _ACC_DTYPE[features.dtype]
    assert margins.dtype == acc and margins.numel() == n
    labels = labels.contiguous()
    if labels.dtype != torch.float32:
        labels = labels.to(torch.float32)
    mask = _prep_mask(mask, features.device)
    dev = features.device
    loss_count = torch.zeros(2, dtype=torch.float64, device=dev)
    mult = torch.empty(n, dtype=acc, device=dev)
    if need_grad:
        grad = torch.empty(d, dtype=acc, device=dev)
        n_rb = int(lib.agd_dense_rowblocks(n, d, a_dtype))
        part = torch.empty(n_rb * d, dtype=acc, device=dev) if n_rb > 1 else grad
    else:
        grad = part = None
        n_rb = 1
    sw = _prep_weights(sample_weight, features.device)
    rc = lib.agd_dense_eval(
        _ptr(features), a_dtype, _ptr(labels), _ptr(mask), _ptr(sw), None,
        n, d, _ptr(grad), _ptr(loss_count), _ptr(margins.contiguous()), _ptr(mult),
        _ptr(part), n_rb, loss_type, 1, 1 if need_grad else 0,
        1, int(os.environ.get("SPARKAGD_NT_LOADS", "1")),
        2, _ptr(_red_ws(features.device)), _stream(features),
    )
    _check(rc)
    return grad, loss_count


def dense_multiplier_loss(
    features: torch.Tensor,
    margins: torch.Tensor,
    labels: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """(mult, loss_count) from precomputed margins — the Gram solver's
    n-space evaluation (no data pass at all)."""
    lib = load()
    n, d = features.shape
    a_dtype = _DTYPE_CODE[features.dtype]
    acc = _ACC_DTYPE[features.dtype]
    assert margins.dtype == acc and margins.numel() == n
    labels = labels.contiguous()
    if labels.dtype != torch.float32:
        labels = labels.to(torch.float32)
    mask = _prep_mask(mask, features.device)
    dev = features.device
    loss_count = torch.zeros(2, dtype=torch.float64, device=dev)
    mult = torch.empty(n, dtype=acc, device=dev)
    sw = _prep_weights(sample_weight, features.device)
    rc = lib.agd_dense_eval(
        _ptr(features), a_dtype, _ptr(labels), _ptr(mask), _ptr(sw), None,
        n, d, None, _ptr(loss_count), _ptr(margins.contiguous()), _ptr(mult),
        None, 1, loss_type, 1, 0,
        1, int(os.environ.get("SPARKAGD_NT_LOADS", "1")),
        2, _ptr(_red_ws(dev)), _stream(features),
    )
    _check(rc)
    return mult, loss_count


def dense_grad_from_mult(features: torch.Tensor, mult: torch.Tensor) -> torch.Tensor:
    """A^T @ mult for an arbitrary multiplier vector (Gram solver's final
    weight materialization)."""
    lib = load()
    n, d = features.shape
    a_dtype = _DTYPE_CODE[features.dtype]
    acc = _ACC_DTYPE[features.dtype]
    assert mult.dtype == acc and mult.numel() == n
    dev = features.device
    grad = torch.empty(d, dtype=acc, device=dev)
    n_rb = int(lib.agd_dense_rowblocks(n, d, a_dtype))
    part = torch.empty(n_rb * d, dtype=acc, device=dev) if n_rb > 1 else grad
    rc = lib.agd_dense_eval(
        _ptr(features), a_dtype, None, None, None, None,
        n, d, _ptr(grad), None, None, _ptr(mult.contiguous()),
        _ptr(part), n_rb, 0, 1, 1,
        1, int(os.environ.get("SPARKAGD_NT_LOADS", "1")),
        3, _ptr(_red_ws(dev)), _stream(features),
    )
    _check(rc)
    return grad


def _csc_grad_skew(csc, csc_heavy, mult: torch.Tensor, d: int) -> torch.Tensor:
    """Deterministic skew-robust CSC gradient: light thread-per-column pass
    for columns <= heavy_T nnz + wave-per-task partials for split heavy
    columns, combined in task order (agd_csc_grad_skew)."""
    lib = load()
    colptr, crow, cval = csc
    grad = torch.empty(d, dtype=torch.float32, device=cval.device)
    rc = lib.agd_csc_grad_skew(
        _ptr(colptr), _ptr(crow), _ptr(cval), _ptr(mult.contiguous()), d,
        int(csc_heavy["heavy_T"]), _ptr(csc_heavy["cols"]),
        _ptr(csc_heavy["taskptr"]), _ptr(csc_heavy["task_idx"]),
        csc_heavy["cols"].numel(), csc_heavy["task_idx"].numel(),
        int(csc_heavy["S"]), _ptr(csc_heavy["partial"]),
        _ptr(csc_heavy.get("order")), _ptr(grad),
        _stream(cval))
    _check(rc)
    return grad


def csr_eval(
    rowptr: torch.Tensor,
    col: torch.Tensor,
    val: torch.Tensor,
    labels: torch.Tensor,
    w: torch.Tensor,
    loss_type: int,
    mask: Optional[torch.Tensor] = None,
    d: Optional[int] = None,
    csc: Optional[Tuple[torch.Tensor, torch.Tensor, torch.Tensor]] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
    csc_heavy: Optional[dict] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    lib = load()
    assert val.is_cuda and val.dtype == torch.float32
    if w.dtype != torch.float32:
        raise TypeError("CSR path requires float32 weights")
    n = rowptr.numel() - 1
    d = d if d is not None else w.numel()
    rowptr = rowptr.contiguous()
    col = col.contiguous()
    if rowptr.dtype != torch.int32:
        rowptr = rowptr.to(torch.int32)
    if col.dtype != torch.int32:
        col = col.to(torch.int32)
    labels = labels.contiguous()
    if labels.dtype != torch.float32:
        labels = labels.to(torch.float32)
    mask = _prep_mask(mask, val.device)

    dev = val.device
    if not need_grad:
        grad = None
        cp = cr = cv = None
    elif csc is None:
        grad = torch.zeros(d, dtype=torch.float32, device=dev)  # atomic path accumulates
        cp = cr = cv = None
    else:
        grad = torch.empty(d, dtype=torch.float32, device=dev)  # CSC path overwrites
        cp, cr, cv = csc
    loss_count = torch.zeros(2, dtype=torch.float64, device=dev)
    margins = torch.empty(n, dtype=torch.float32, device=dev)
    mult = torch.empty(n, dtype=torch.float32, device=dev)

    sw = _prep_weights(sample_weight, val.device)
    # skewed columns: let the C call stop after the multiplier, then run the
    # skew-robust gradient on the mult buffer
    skew = need_grad and csc is not None and csc_heavy is not None
    rc = lib.agd_csr_eval(
        _ptr(rowptr), _ptr(col), _ptr(val), _ptr(labels), _ptr(mask), _ptr(sw),
        _ptr(w.contiguous()), n, val.numel(), d, _ptr(grad), _ptr(loss_count),
        _ptr(margins), _ptr(mult), loss_type,
        _ptr(cp), _ptr(cr), _ptr(cv), 1 if (need_grad and not skew) else 0, 0,
        _ptr(_red_ws(val.device)), _stream(val),
    )
    _check(rc)
    if skew:
        grad = _csc_grad_skew(csc, csc_heavy, mult, d)
    return grad, loss_count


def dense_margins_multi(features: torch.Tensor, wflat: torch.Tensor, k: int,
                        kc: int) -> torch.Tensor:
    """Padded flat margins [n*KC] = A @ pad(W [d,K] -> [d,KC])."""
    lib = load()
    n, d = features.shape
    a_dtype = _DTYPE_CODE[features.dtype]
    if features.dtype == torch.float64:
        raise NotImplementedError("multi-class GPU path supports bf16/f32/f8 shards")
    dev = features.device
    w2 = wflat.reshape(d, k).to(torch.float32)
    if kc != k:
        wp = torch.zeros((d, kc), dtype=torch.float32, device=dev)
        wp[:, :k] = w2
    else:
        wp = w2.contiguous()
    Z = torch.empty(n * kc, dtype=torch.float32, device=dev)
    rc = lib.agd_margins_multi(_ptr(features), a_dtype, _ptr(wp), n, d, kc,
                               _ptr(Z), _stream(features))
    _check(rc)
    return Z


def multiplier_multi(
    margins_padded_flat: torch.Tensor,
    labels: torch.Tensor,
    k: int,
    kc: int,
    mask: Optional[torch.Tensor] = None,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """(M [n*kc] f32, loss_count f64[2]) from padded softmax margins —
    the n-space multiplier stage shared by the dense and CSR paths."""
    lib = load()
    dev = margins_padded_flat.device
    n = margins_padded_flat.numel() // kc
    labels = labels.contiguous()
    if labels.dtype != torch.float32:
        labels = labels.to(torch.float32)
    mask = _prep_mask(mask, dev)
    sw = _prep_weights(sample_weight, dev)
    M = torch.empty(n * kc, dtype=torch.float32, device=dev)
    loss_count = torch.zeros(2, dtype=torch.float64, device=dev)
    rc = lib.agd_multiplier_multi(_ptr(margins_padded_flat.contiguous()),
                                  _ptr(labels), _ptr(mask), _ptr(sw), n, k, kc,
                                  _ptr(M), _ptr(loss_count),
                                  _ptr(_red_ws(dev)), _stream(margins_padded_flat))
    _check(rc)
    return M, loss_count


def dense_eval_multi_from_margins(
    features: torch.Tensor,
    margins_padded_flat: torch.Tensor,
    labels: torch.Tensor,
    k: int,
    kc: int,
    mask: Optional[torch.Tensor] = None,
    need_grad: bool = True,
    sample_weight: Optional[torch.Tensor] = None,
) -> Tuple[Optional[torch.Tensor], torch.Tensor]:
    lib = load()
    n, d = features.shape
    a_dtype = _DTYPE_CODE[features.dtype]
    dev = features.device
    M, loss_count = multiplier_multi(margins_padded_flat, labels, k, kc, mask,
                                     sample_weight)
    if not need_grad:
        return None, loss_count
    algo = os.environ.get("SPARKAGD_MULTI_GRAD", "auto")
    if features.dtype == torch.bfloat16 and algo in ("auto", "gemm"):
        # grad = A^T·M as a skinny hipBLASLt TN GEMM — the per-element KC-fma
        # VALU kernel is issue-bound ~3.5x the A-stream floor
        # (profiles/r01_multiclass_trace_gemm.txt). Multipliers round to bf16
        # (standard mixed precision; A is bf16, accumulation f32);
        # SPARKAGD_MULTI_GRAD=valu selects the exact-f32-multiplier kernel.
        gradp = gemm_bf16f32_tn(features, M.reshape(n, kc)).reshape(-1)
    else:
        n_rb = int(lib.agd_multi_rowblocks(n, d, kc))
        gradp = torch.empty(d * kc, dtype=torch.float32, device=dev)
        part = torch.empty(n_rb * d * kc, dtype=torch.float32, device=dev) if n_rb > 1 else gradp
        rc = lib.agd_grad_multi(_ptr(features), a_dtype, _ptr(M), n, d, kc,
                                _ptr(part), n_rb, _ptr(gradp), _stream(features))
        _check(rc)
    if kc != k:
        grad = gradp.reshape(d, kc)[:, :k].reshape(-1).contiguous()
    else:
        grad = gradp
    return grad, loss_count


def csr_margins_multi(rowptr, col, val, wflat: torch.Tensor, k: int,
                      kc: int, d: int) -> torch.Tensor:
    """Padded flat margins [n*KC] = CSR(A) @ pad(W [d,K] -> [d,KC]).

    SPARKAGD_CSR_MULTI_W=bf16 gathers bf16-rounded W rows (half the gather
    bytes, double the effective LLC coverage of W — the pass is
    LLC-miss-bound once W exceeds the last-level cache); default f32 keeps
    exact weights (the CSR values are f32)."""
    lib = load()
    n = rowptr.numel() - 1
    dev = val.device
    wdt = (torch.bfloat16
           if os.environ.get("SPARKAGD_CSR_MULTI_W", "f32") == "bf16"
           else torch.float32)
    w2 = wflat.reshape(d, k).to(wdt)
    if kc != k:
        wp = torch.zeros((d, kc), dtype=wdt, device=dev)
        wp[:, :k] = w2
    else:
        wp = w2.contiguous()
    Z = torch.empty(n * kc, dtype=torch.float32, device=dev)
    rowptr, col = _csr_idx(rowptr, col)
    rc = lib.agd_csr_margins_multi(_ptr(rowptr), _ptr(col),
                                   _ptr(val.contiguous()), _ptr(wp), n, kc,
                                   0 if wdt == torch.bfloat16 else 1,
                                   _ptr(Z), _stream(val))
    _check(rc)
    return Z


def csc_grad_multi(colptr, row, cval, M: torch.Tensor, d: int,
                   kc: int, csc_heavy: Optional[dict] = None) -> torch.Tensor:
    """grad flat [d*KC] = A^T·M via the deterministic CSC gather; skewed
    columns (csc_heavy, built by CSRShard) run the wave-per-task split."""
    lib = load()
    grad = torch.empty(d * kc, dtype=torch.float32, device=cval.device)
    if csc_heavy is not None:
        n_tasks = csc_heavy["task_idx"].numel()
        partial = (torch.empty(n_tasks * kc, dtype=torch.float32,
                               device=cval.device) if n_tasks > 0 else None)
        rc = lib.agd_csc_grad_multi(
            _ptr(colptr.contiguous()), _ptr(row.contiguous()),
            _ptr(cval.contiguous()), _ptr(M.contiguous()), d, kc, _ptr(grad),
            int(csc_heavy["heavy_T"]), _ptr(csc_heavy["cols"]),
            _ptr(csc_heavy["taskptr"]), _ptr(csc_heavy["task_idx"]),
            csc_heavy["cols"].numel(), n_tasks, int(csc_heavy["S"]),
            _ptr(partial), _ptr(csc_heavy.get("order")), _stream(cval))
    else:
        rc = lib.agd_csc_grad_multi(
            _ptr(colptr.contiguous()), _ptr(row.contiguous()),
            _ptr(cval.contiguous()), _ptr(M.contiguous()), d, kc, _ptr(grad),
            0, None, None, None, 0, 0, 0, None, None, _stream(cval))
    _check(rc)
    return grad


def gemm_bf16f32_nt(A: torch.Tensor, B: torch.Tensor, C: torch.Tensor,
                    beta: float = 0.0) -> torch.Tensor:
    """C[m,n] (f32) = A[m,k] (bf16) @ B[n,k]^T (bf16) + beta*C via hipBLASLt
    (fp32 accumulation, fp32 output — used by the Gram-operator build)."""
    lib = load()
    assert A.dtype == torch.bfloat16 and B.dtype == torch.bfloat16
    assert C.dtype == torch.float32
    m, k = A.shape
    n, k2 = B.shape
    assert k == k2 and C.shape == (m, n)
    assert A.is_contiguous() and B.is_contiguous() and C.is_contiguous()
    rc = lib.agd_gemm_bf16f32_nt(_ptr(A), _ptr(B), _ptr(C), m, n, k,
                                 float(beta), _stream(A))
    _check(rc)
    return C


def gemm_bf16f32_tn(A: torch.Tensor, M: torch.Tensor) -> torch.Tensor:
    """grad[d,kc] (f32) = A[n,d]^T (bf16) @ M[n,kc] (bf16-rounded) via
    hipBLASLt — the multinomial gradient GEMM (fp32 accumulation). Accepts M
    in f32 (rounded here) or bf16; works for any kc."""
    lib = load()
    assert A.dtype == torch.bfloat16
    n, d = A.shape
    n2, kc = M.shape
    assert n == n2
    if M.dtype != torch.bfloat16:
        M = M.to(torch.bfloat16)
    M = M.contiguous()
    grad = torch.empty((d, kc), dtype=torch.float32, device=A.device)
    rc = lib.agd_gemm_bf16f32_tn(_ptr(A.contiguous()), _ptr(M), _ptr(grad),
                                 n, d, kc, _stream(A))
    _check(rc)
    return grad


def _csr_idx(rowptr: torch.Tensor, col: torch.Tensor):
    """Kernel index arrays are int32: convert int64 inputs (no-op otherwise).
    Passing int64 straight through would silently misread (the bug the
    MixedShard GPU test caught in round 2)."""
    rowptr = rowptr.contiguous()
    col = col.contiguous()
    if rowptr.dtype != torch.int32:
        rowptr = rowptr.to(torch.int32)
    if col.dtype != torch.int32:
        col = col.to(torch.int32)
    return rowptr, col


def csr_margins(rowptr, col, val, v: torch.Tensor) -> torch.Tensor:
    lib = load()
    n = rowptr.numel() - 1
    rowptr, col = _csr_idx(rowptr, col)
    margins = torch.empty(n, dtype=torch.float32, device=val.device)
    rc = lib.agd_csr_eval(
        _ptr(rowptr), _ptr(col), _ptr(val.contiguous()),
        None, None, None, _ptr(v.contiguous()), n, val.numel(), v.numel(),
        None, None, _ptr(margins), None, 0, None, None, None, 0, 1,
        _ptr(_red_ws(val.device)), _stream(val),
    )
    _check(rc)
    return margins


def csr_eval_from_margins(rowptr, col, val, margins, labels, loss_type,
                          mask=None, d=None, csc=None, need_grad=True,
                          sample_weight=None, csc_heavy=None):
    lib = load()
    n = rowptr.numel() - 1
    rowptr, col = _csr_idx(rowptr, col)
    d = d if d is not None else 0
    labels = labels.contiguous()
    if labels.dtype != torch.float32:
        labels = labels.to(torch.float32)
    mask = _prep_mask(mask, val.device)
    dev = val.device
    if not need_grad:
        grad = None; cp = cr = cv = None
    elif csc is None:
        grad = torch.zeros(d, dtype=torch.float32, device=dev); cp = cr = cv = None
    else:
        grad = torch.empty(d, dtype=torch.float32, device=dev); cp, cr, cv = csc
    loss_count = torch.zeros(2, dtype=torch.float64, device=dev)
    mult = torch.empty(n, dtype=torch.float32, device=dev)
    sw = _prep_weights(sample_weight, val.device)
    skew = need_grad and csc is not None and csc_heavy is not None
    rc = lib.agd_csr_eval(
        _ptr(rowptr), _ptr(col), _ptr(val.contiguous()),
        _ptr(labels), _ptr(mask), _ptr(sw), None, n, val.numel(), d,
        _ptr(grad), _ptr(loss_count), _ptr(margins.contiguous()), _ptr(mult),
        loss_type, _ptr(cp), _ptr(cr), _ptr(cv),
        1 if (need_grad and not skew) else 0, 2,
        _ptr(_red_ws(val.device)), _stream(val),
    )
    _check(rc)
    if skew:
        grad = _csc_grad_skew(csc, csc_heavy, mult, d)
    return grad, loss_count


_VEC_DTYPE = {torch.float32: 1, torch.float64: 2}


def axpby(a: float, x: torch.Tensor, b: float, y: torch.Tensor, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    lib = load()
    if out is None:
        out = torch.empty_like(x)
    rc = lib.agd_axpby(float(a), _ptr(x.contiguous()), float(b), _ptr(y.contiguous()),
                       _ptr(out), x.numel(), _VEC_DTYPE[x.dtype], _stream(x))
    _check(rc)
    return out


def prox(kind: int, w: torch.Tensor, g: torch.Tensor, step: float, lam: float,
         lam2: float = 0.0) -> Tuple[torch.Tensor, torch.Tensor]:
    lib = load()
    out = torch.empty_like(w)
    reg = torch.zeros((), dtype=torch.float64, device=w.device)
    rc = lib.agd_prox(kind, _ptr(w.contiguous()), _ptr(g.contiguous()), float(step),
                      float(lam), float(lam2), _ptr(out), _ptr(reg), w.numel(),
                      _VEC_DTYPE[w.dtype], _ptr(_red_ws(w.device)), _stream(w))
    _check(rc)
    return out, reg


def fused_scalars(x: torch.Tensor, y: torch.Tensor, g_y: torch.Tensor, x_old: torch.Tensor) -> torch.Tensor:
    lib = load()
    out = torch.zeros(5, dtype=torch.float64, device=x.device)
    rc = lib.agd_fused_scalars(_ptr(x.contiguous()), _ptr(y.contiguous()),
                               _ptr(g_y.contiguous()), _ptr(x_old.contiguous()),
                               _ptr(out), x.numel(), _VEC_DTYPE[x.dtype],
                               _ptr(_red_ws(x.device)), _stream(x))
    _check(rc)
    return out


def dot_diff(x: torch.Tensor, y: torch.Tensor, g_x: torch.Tensor, g_y: torch.Tensor) -> torch.Tensor:
    lib = load()
    out = torch.zeros((), dtype=torch.float64, device=x.device)
    rc = lib.agd_dot_diff(_ptr(x.contiguous()), _ptr(y.contiguous()),
                          _ptr(g_x.contiguous()), _ptr(g_y.contiguous()),
                          _ptr(out), x.numel(), _VEC_DTYPE[x.dtype],
                          _ptr(_red_ws(x.device)), _stream(x))
    _check(rc)
    return out


def gram_mult_affine(xm: torch.Tensor, zm: torch.Tensor, a: float, b: float,
                     labels: torch.Tensor, loss_type: int,
                     sample_weight: Optional[torch.Tensor] = None,
                     mult_out: Optional[torch.Tensor] = None,
                     lc_out: Optional[torch.Tensor] = None):
    """(mult, loss_count) at margins a*xm + b*zm without materializing the
    combination — the Gram solver's fused y-evaluation (binary losses,
    full batch)."""
    lib = load()
    n = xm.numel()
    dev = xm.device
    labels = labels.contiguous()
    if labels.dtype != torch.float32:
        labels = labels.to(torch.float32)
    mult = mult_out if mult_out is not None else torch.empty(
        n, dtype=torch.float32, device=dev)
    lc = lc_out if lc_out is not None else torch.empty(
        2, dtype=torch.float64, device=dev)
    lc.zero_()
    sw = _prep_weights(sample_weight, dev)
    rc = lib.agd_gram_mult_affine(_ptr(xm.contiguous()), _ptr(zm.contiguous()),
                                  float(a), float(b), _ptr(labels), _ptr(sw),
                                  loss_type, n, _ptr(mult), _ptr(lc),
                                  _ptr(_red_ws(dev)), _stream(xm))
    _check(rc)
    return mult, lc


def gram_state_update(gm_raw: torch.Tensor, m_y: torch.Tensor,
                      xm_old: torch.Tensor, zm_old: torch.Tensor,
                      inv_c: float, theta: float, pz: float, pg: float,
                      xb_t: torch.Tensor, md: torch.Tensor,
                      mstore_t: torch.Tensor, zm_new: torch.Tensor,
                      xm_new: torch.Tensor) -> None:
    """One fused pass finishing a Gram basis registration + the AT margin
    updates (see k_gram_state_update)."""
    lib = load()
    rc = lib.agd_gram_state_update(
        _ptr(gm_raw.contiguous()), _ptr(m_y.contiguous()),
        _ptr(xm_old.contiguous()), _ptr(zm_old.contiguous()),
        float(inv_c), float(theta), float(pz), float(pg), gm_raw.numel(),
        _ptr(xb_t), _ptr(md), _ptr(mstore_t), _ptr(zm_new), _ptr(xm_new),
        _stream(gm_raw))
    _check(rc)


def at_margin_update(zm_old: torch.Tensor, xm_old: torch.Tensor,
                     gm: torch.Tensor, pz: float, pg: float, theta: float):
    """(zm_new, xm_new) = (pz*zm + pg*gm, (1-theta)*xm + theta*zm_new) in one
    fused pass (direct tracked path)."""
    lib = load()
    zm_new = torch.empty_like(zm_old)
    xm_new = torch.empty_like(xm_old)
    rc = lib.agd_at_margin_update(
        _ptr(zm_old.contiguous()), _ptr(xm_old.contiguous()),
        _ptr(gm.contiguous()), float(pz), float(pg), float(theta),
        zm_old.numel(), _VEC_DTYPE[zm_old.dtype], _ptr(zm_new), _ptr(xm_new),
        _stream(zm_old))
    _check(rc)
    return zm_new, xm_new
