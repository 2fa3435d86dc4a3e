"""Host-streamed shards: train on data larger than HBM.

The reference inherits Spark's cached-RDD spill/recompute path for data that
exceeds executor memory (SURVEY.md §5 'Failure detection' — implicit, free).
The MI355X-native analog: a shard whose features live in PINNED HOST MEMORY
and stream through a pair of device chunk buffers, with the H2D copies on a
dedicated HIP copy stream overlapped against the evaluation kernels on the
compute stream (double buffering: copy chunk i+1 while chunk i computes).

The evaluation is PCIe-bound by construction (~tens of GB/s vs the ~6.6 TB/s
HBM path), so this is a CAPACITY escape hatch — data up to host-RAM size per
GPU — not a speed path; the overlap keeps it at the copy ceiling instead of
copy+compute serialized. Labels, margins and the weight vector stay
device-resident, so margin-state tracking and the n-space multiplier work
exactly as for in-HBM shards.

Binary losses only (the multiclass path dispatches on dense/CSR shard kinds).
On CPU the same code degrades to a plain chunked loop (tests/CI tier).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import ops


class HostStreamedDenseShard:
    """Row-sharded dense design matrix in pinned host memory, streamed
    through device chunk buffers with copy/compute overlap."""

    kind = "dense_streamed"

    def __init__(
        self,
        features_host: torch.Tensor,
        labels: torch.Tensor,
        device: torch.device | str = "cuda",
        chunk_rows: int = 8192,
        sample_weight: Optional[torch.Tensor] = None,
    ):
        if features_host.ndim != 2:
            raise ValueError("features must be [n, d]")
        self.device = torch.device(device)
        self._cuda = self.device.type == "cuda"
        feats = features_host.contiguous()
        if self._cuda and not feats.is_pinned():
            feats = feats.pin_memory()
        self.features_host = feats
        self.labels = labels.to(self.device)
        if self.labels.dtype not in (torch.float32, torch.float64):
            self.labels = self.labels.to(torch.float32)
        self.sample_weight = None
        if sample_weight is not None:
            self.sample_weight = sample_weight.to(self.device, torch.float32)
        self.chunk_rows = int(chunk_rows)
        n, d = feats.shape
        self._n, self._d = int(n), int(d)
        if self._cuda:
            self._buf = [
                torch.empty((min(self.chunk_rows, self._n), d),
                            dtype=feats.dtype, device=self.device)
                for _ in range(2)
            ]
            self._copy_stream = torch.cuda.Stream(self.device)
            self._copy_done = [torch.cuda.Event(), torch.cuda.Event()]
            self._compute_done = [torch.cuda.Event(), torch.cuda.Event()]

    # --- shard interface ---

    @property
    def n(self) -> int:
        return self._n

    @property
    def d(self) -> int:
        return self._d

    @property
    def nbytes(self) -> int:
        return self.features_host.numel() * self.features_host.element_size()

    def _chunks(self):
        for lo in range(0, self._n, self.chunk_rows):
            yield lo, min(lo + self.chunk_rows, self._n)

    def _for_each_chunk(self, fn) -> None:
        """Run ``fn(chunk_features_device, lo, hi)`` over all chunks with
        double-buffered H2D copies overlapping the compute stream."""
        if not self._cuda:
            for lo, hi in self._chunks():
                fn(self.features_host[lo:hi], lo, hi)
            return
        main = torch.cuda.current_stream(self.device)
        chunks = list(self._chunks())
        for i, (lo, hi) in enumerate(chunks):
            b = i % 2
            buf = self._buf[b][: hi - lo]
            with torch.cuda.stream(self._copy_stream):
                # Buffer reuse: wait for the LAST compute that read this
                # buffer — chunk i-2 of this pass, or (for i < 2) the tail
                # chunks of the PREVIOUS pass, whose events persist across
                # calls. Waiting unconditionally covers back-to-back passes
                # with no intervening host sync (an event never recorded is
                # a no-op wait).
                self._copy_stream.wait_event(self._compute_done[b])
                buf.copy_(self.features_host[lo:hi], non_blocking=True)
                self._copy_done[b].record(self._copy_stream)
            main.wait_event(self._copy_done[b])
            fn(buf, lo, hi)
            self._compute_done[b].record(main)

    def _slice(self, t: Optional[torch.Tensor], lo: int, hi: int):
        return None if t is None else t[lo:hi]

    def eval(self, w: torch.Tensor, loss_type: int,
             mask: Optional[torch.Tensor] = None, need_grad: bool = True
             ) -> Tuple[Optional[torch.Tensor], torch.Tensor]:
        grad = None
        loss_count = torch.zeros(2, dtype=torch.float64, device=self.device)

        def step(buf, lo, hi):
            nonlocal grad
            g, lc = ops.dense_eval(buf, self.labels[lo:hi], w, loss_type,
                                   self._slice(mask, lo, hi), need_grad,
                                   self._slice(self.sample_weight, lo, hi))
            loss_count.add_(lc)
            if need_grad:
                grad = g if grad is None else ops.axpby(1.0, grad, 1.0, g)

        self._for_each_chunk(step)
        return grad, loss_count

    def margins(self, v: torch.Tensor) -> torch.Tensor:
        out = torch.empty(self._n, dtype=v.dtype, device=self.device)

        def step(buf, lo, hi):
            out[lo:hi] = ops.dense_margins(buf, v)

        self._for_each_chunk(step)
        return out

    def eval_from_margins(self, margins: torch.Tensor, loss_type: int,
                          mask: Optional[torch.Tensor] = None,
                          need_grad: bool = True):
        # The multiplier stage is n-space elementwise (no feature pass);
        # the torch formulation runs on the device margins directly.
        mult, loss = ops.reference._multiplier_and_loss(margins, self.labels,
                                                        loss_type)
        mult, loss, count = ops.reference._apply_mask_weight(
            mult, loss, mask, self.sample_weight, self._n, self.device)
        loss_count = torch.stack([loss.to(torch.float64).sum(), count])
        if not need_grad:
            return None, loss_count
        grad = None

        def step(buf, lo, hi):
            nonlocal grad
            g = ops.dense_grad_from_mult(buf, mult[lo:hi].contiguous())
            grad = g if grad is None else ops.axpby(1.0, grad, 1.0, g)

        self._for_each_chunk(step)
        return grad, loss_count
