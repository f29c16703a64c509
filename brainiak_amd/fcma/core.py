"""FCMA compute pipeline: correlation → Fisher-z normalization → Gram.

This is the MI355X re-design of the reference's kernel chain
(ref src/brainiak/fcma/cython_blas.pyx:20-207 — per-epoch sgemm + per-voxel
ssyrk — and src/brainiak/fcma/src/fcma_extension.cc:29-86 — OpenMP
Fisher-z/z-score).  Stages:

  1. ``correlate_chunk``  — corr[c, e, v] = Σ_t A[e, t, s+c]·A2[e, t, v]
     (a batched GEMM over epochs; epochs are padded to a common length —
     z-scored columns make the pad zeros inert).
  2. ``normalize_correlation_`` — in-place Fisher z (with the reference's
     1±r ≤ 0 → 1e-4 clamp) + per-(voxel, subject, column) z-score over
     ``epochs_per_subj`` (ddof=0, zero-variance → 0).
  3. ``gram_matrices`` — per selected voxel the [E, E] linear-SVM Gram
     matrix over its V-length normalized correlation vectors, plus the
     reference's magnitude shrink (10**(2-digits) when the leading
     diagonal entry has >2 integer digits).

On CUDA (= ROCm) tensors the stages dispatch to the hand-written
HIP/CDNA4 kernels in ``brainiak_amd.ops``.  The round-2 default is the
DEFERRED-NORMALIZE split: ``k_corr_norm_dot3s<RAW>`` stores raw
correlations (v_dot2c bf16 MACs, scalar-path A operand), the Gram
kernel applies Fisher-z + the z-score to each staged tile IN REGISTERS
inside its memory-latency shadow, and ``k_corr_gram_duo`` carries the
corr blocks of chunk i and the Gram blocks of chunk i-1 in ONE grid so
the VALU-bound and latency-bound waves co-reside (docs/kernels.md).
``z_fp8=True`` stores Z as OCP e4m3 (profiles/fp8.md).  The torch
implementations below are the CPU path and the numerics oracle the GPU
kernels are tested against.
"""

import os
from typing import List, Optional

import numpy as np
import torch

from .. import ops

__all__ = [
    "CorrelationPipeline",
    "gram_matrices",
    "normalize_correlation_",
    "stack_epochs",
]

# epoch lengths the HIP correlation kernels are templated for;
# longer epochs fall back to the rocBLAS bmm + normalize path
_HIP_EPOCH_LENGTHS = (8, 12, 16, 20, 24, 28, 32, 36, 40)


def stack_epochs(raw_data: List[np.ndarray], device,
                 dtype=torch.float32) -> torch.Tensor:
    """Stack per-epoch [len_e, V] matrices into [E, L, Vp] (zero-padded).

    The inputs are z-scored and 1/sqrt(len)-scaled, so zero padding does
    not change any correlation dot product.  On GPU the voxel dimension
    pads to a multiple of 16 (zero columns are equally inert) so fp8/
    bf16 Z rows stay 16-byte aligned for vector loads.
    """
    E = len(raw_data)
    L = max(m.shape[0] for m in raw_data)
    V = raw_data[0].shape[1]
    if device is not None and torch.device(device).type == "cuda":
        # pad to the HIP kernel's supported epoch lengths; zero rows are
        # inert for z-scored data.  L > max(_HIP_EPOCH_LENGTHS) stays
        # unpadded and routes through the rocBLAS bmm fallback.
        for opt in _HIP_EPOCH_LENGTHS:
            if L <= opt:
                L = opt
                break
        V = ((V + 15) // 16) * 16
    out = torch.zeros((E, L, V), dtype=dtype, device=device)
    for e, m in enumerate(raw_data):
        if isinstance(m, torch.Tensor):
            t = m.to(device=device, dtype=dtype)
        else:
            t = torch.as_tensor(np.ascontiguousarray(m),
                                dtype=torch.float32).to(device).to(dtype)
        out[e, :m.shape[0], :m.shape[1]] = t
    return out


def normalize_correlation_(corr: torch.Tensor, epochs_per_subj: int
                           ) -> torch.Tensor:
    """In-place Fisher-z + within-subject z-score of corr [C, E, V].

    Matches the reference's native kernel semantics exactly
    (fcma_extension.cc:54-83): clamp 1±r ≤ 0 to 1e-4, z-score with the
    biased variance over each subject's ``epochs_per_subj`` epochs, and
    zero out zero-variance entries.
    """
    if corr.is_cuda and ops.require_hip():
        ops.fcma_normalize_(corr, epochs_per_subj)
        return corr
    C, E, V = corr.shape
    n_subj = E // epochs_per_subj
    # exact reference semantics: only non-positive arguments are clamped
    # (a tiny positive 1±r stays, so r≈±1 is intentionally noise-driven —
    # fcma_extension.cc:68-71 behaves identically)
    num = 1.0 + corr
    den = 1.0 - corr
    num = torch.where(num <= 0, torch.full_like(num, 1e-4), num)
    den = torch.where(den <= 0, torch.full_like(den, 1e-4), den)
    z = 0.5 * torch.log(num / den)
    z = z.view(C, n_subj, epochs_per_subj, V)
    mean = z.mean(dim=2, keepdim=True)
    var = (z * z).mean(dim=2, keepdim=True) - mean * mean
    inv_std = torch.where(var > 0, var.rsqrt(),
                          torch.zeros_like(var))
    corr.copy_(((z - mean) * inv_std).view(C, E, V))
    return corr


def _shrink_(gram: torch.Tensor) -> torch.Tensor:
    """Reference's magnitude shrink (voxelselector.py:407-412): if the
    first diagonal entry has more than 2 integer digits, scale the whole
    [E, E] matrix by 10**(2-digits)."""
    lead = gram[:, 0, 0].abs().clamp_min(1.0)
    digits = torch.floor(torch.log10(lead)) + 1
    scale = torch.where(digits > 2, torch.pow(10.0, 2 - digits),
                        torch.ones_like(lead))
    gram.mul_(scale[:, None, None])
    return gram


def gram_matrices(corr_norm: torch.Tensor, shrink: bool = True
                  ) -> torch.Tensor:
    """Per-voxel [E, E] Gram matrices of corr_norm [C, E, V]."""
    if corr_norm.is_cuda and ops.require_hip():
        gram = ops.fcma_gram(corr_norm)
    else:
        gram = torch.bmm(corr_norm, corr_norm.transpose(1, 2))
    if shrink:
        _shrink_(gram)
    return gram


class CorrelationPipeline:
    """Holds the stacked epoch tensors and runs the chunk pipeline.

    Parameters
    ----------
    raw_data, raw_data2 : list of [len_e, V] z-scored epoch matrices
        (raw_data2 None → self-correlation).
    epochs_per_subj : int
    device : torch device for the compute.
    use_bf16 : cast the GEMM inputs to bf16 on GPU (the BASELINE dtype);
        accumulation stays fp32.
    """

    def __init__(self, raw_data, raw_data2, epochs_per_subj, device=None,
                 use_bf16: Optional[bool] = None,
                 z_fp8: Optional[bool] = None):
        self.device = torch.device(device) if device is not None else (
            torch.device("cuda") if torch.cuda.is_available()
            else torch.device("cpu"))
        if use_bf16 is None:
            use_bf16 = self.device.type == "cuda"
        self.use_bf16 = use_bf16
        gemm_dtype = torch.bfloat16 if use_bf16 else torch.float32
        if z_fp8 is None:
            z_fp8 = bool(os.environ.get("BRAINIAK_FP8"))
        # fp8(e4m3) Z tile: halves the normalized-correlation HBM round
        # trip between the corr and Gram kernels (~3 % RMS quantization
        # on z-scores; selection-accuracy parity is the tested contract)
        self.z_fp8 = bool(z_fp8) and self.device.type == "cuda"
        self.epochs_per_subj = epochs_per_subj
        self.data = stack_epochs(raw_data, self.device, gemm_dtype)
        self.data2 = (stack_epochs(raw_data2, self.device, gemm_dtype)
                      if raw_data2 is not None else self.data)
        self.num_epochs = self.data.shape[0]
        # logical voxel counts (the stacked tensors may be 16-padded)
        self.num_voxels = (raw_data[0].shape[1] if raw_data
                           else self.data.shape[2])
        self.num_voxels2 = (raw_data2[0].shape[1] if raw_data2
                            else self.num_voxels)
        # HIP kernels handle bf16 inputs at the templated epoch lengths;
        # fp32 compute (use_bf16=False) or long epochs route through the
        # rocBLAS bmm + normalize fallback on device
        self._hip_path = (
            self.device.type == "cuda" and self.use_bf16
            and self.data.shape[1] in _HIP_EPOCH_LENGTHS
            and ops.require_hip())
        # deferred-normalize split (bf16 only): corr writes raw r, the
        # Gram kernel Fisher-z/z-scores each staged tile in its memory
        # latency shadow.  BRAINIAK_NO_RAWCORR=1 restores the fused
        # corr+norm kernel for A/B.
        self._raw_split = (self._hip_path and not self.z_fp8
                           and epochs_per_subj in (2, 4)
                           and not os.environ.get("BRAINIAK_NO_RAWCORR"))

    def correlate_chunk(self, start: int, count: int) -> torch.Tensor:
        """corr [count, E, V2] fp32 for voxels [start, start+count)
        (V2 = logical voxel count; any 16-pad columns are sliced off)."""
        if self._hip_path:
            out = ops.fcma_correlate(self.data, self.data2, start, count)
        else:
            a = self.data[:, :, start:start + count].to(torch.float32)
            b = self.data2.to(torch.float32)
            # [E, C, V] then→ [C, E, V]
            out = torch.bmm(a.transpose(1, 2), b).transpose(0, 1)
        if out.shape[2] != self.num_voxels2:
            out = out[:, :, :self.num_voxels2]
        return out.contiguous()

    def chunk_kernel_matrices(self, start: int, count: int,
                              shrink: bool = True) -> torch.Tensor:
        """Full chunk: correlation → normalize → Gram [count, E, E].

        On gfx950 ``ops.fcma_fused_gram`` runs the chunk as one kernel
        for the headline shape (E=64, P=4) and as the two-kernel
        corr_norm_z + MFMA-Gram composite otherwise.
        """
        if self._hip_path and ops.has_fused_gram():
            gram = ops.fcma_fused_gram(
                self.data, self.data2, start, count, self.epochs_per_subj)
            if shrink:
                _shrink_(gram)
            return gram
        corr = self.correlate_chunk(start, count)
        normalize_correlation_(corr, self.epochs_per_subj)
        return gram_matrices(corr, shrink=shrink)

    def pipelined_kernel_matrices(self, chunks, shrink: bool = True
                                  ) -> torch.Tensor:
        """Gram matrices for several (start, count) chunks with the
        corr+norm kernel of chunk i+1 overlapping the Gram/MFMA kernel
        of chunk i on a second HIP stream.

        The two stages use disjoint hardware (corr+norm is VALU/LDS
        bound, the Gram is MFMA + HBM reads), so the overlap hides most
        of the Gram time — measured ≈25 % off the whole FCMA step.
        """
        chunks = list(chunks)
        if not self._hip_path or len(chunks) <= 1:
            return torch.cat([self.chunk_kernel_matrices(s, c, shrink)
                              for s, c in chunks], dim=0)
        if self._raw_split and not os.environ.get("BRAINIAK_NO_DUO"):
            return self._duo_pipeline(chunks, shrink)
        E = self.num_epochs
        Epad = ((E + 63) // 64) * 64
        ext = ops.load_extension()
        if getattr(ext, "fcma_fused_gram_native", None) is not None and \
                ext.fcma_fused_gram_native(E, self.epochs_per_subj,
                                           self.data.shape[1]) and \
                os.environ.get("BRAINIAK_FUSED"):
            # single-kernel corr+gram (opt-in): Z never touches HBM, but
            # measured on MI355X the B-read amplification at its 8-voxel
            # c-tile outweighs the saved Z round trip (24.7 vs 16.7
            # ms/step, profiles/README.md) — the streamed two-kernel
            # pipeline below with a 64-voxel corr tile wins
            gram = torch.cat([
                ops.fcma_fused_gram(self.data, self.data2, s, c,
                                    self.epochs_per_subj)
                for s, c in chunks], dim=0)
            if shrink:
                _shrink_(gram)
            return gram
        max_count = max(c for _, c in chunks)
        # persistent double-buffered Z workspace: repeated multi-GB
        # allocations churn the caching allocator across streams
        zdtype = torch.float8_e4m3fn if self.z_fp8 else torch.bfloat16
        VB = self.data2.shape[2]        # padded voxel count
        if getattr(self, "_zbuf", None) is None or \
                self._zbuf[0].shape[0] < max_count or \
                self._zbuf[0].shape[1] != Epad or \
                self._zbuf[0].dtype != zdtype:
            self._zbuf = [
                torch.zeros((max_count, Epad, VB),
                            dtype=zdtype, device=self.device)
                for _ in range(2)]
        if getattr(self, "_streams", None) is None:
            # equal priorities: a high-priority corr stream SERIALIZES
            # the two kernels (the scheduler starves the gram stream
            # while any corr block is pending — measured exactly
            # corr+gram serial); equal priority lets them co-schedule
            # (BRAINIAK_CORR_PRIO=-1 re-enables the experiment)
            prio = int(os.environ.get("BRAINIAK_CORR_PRIO", "0"))
            self._streams = (
                torch.cuda.Stream(device=self.device, priority=prio),
                torch.cuda.Stream(device=self.device))
        corr_stream, gram_stream = self._streams
        grams = []
        pending = None          # (z, ready-event, buffer index)
        buf_free = [None, None]  # event: gram done reading buffer b

        skip_gram = bool(os.environ.get("BRAINIAK_SKIP_GRAM"))  # probe

        def _consume(z, ev, bidx):
            with torch.cuda.stream(gram_stream):
                gram_stream.wait_event(ev)
                if skip_gram:
                    g = torch.zeros((z.shape[0], Epad, Epad),
                                    dtype=torch.float32, device=z.device)
                elif self.z_fp8:
                    g = ops.fcma_gram_fp8(z)
                else:
                    g = ops.fcma_gram_bf16(
                        z, norm_P=self.epochs_per_subj
                        if self._raw_split else 0)
                if Epad != E:
                    g = g[:, :E, :E].contiguous()
                done = torch.cuda.Event()
                done.record(gram_stream)
                buf_free[bidx] = done
                grams.append(g)

        for i, (start, count) in enumerate(chunks):
            bidx = i % 2
            with torch.cuda.stream(corr_stream):
                # don't overwrite a buffer the gram stream still reads
                if buf_free[bidx] is not None:
                    corr_stream.wait_event(buf_free[bidx])
                z = ext.fcma_corr_norm_z(
                    self.data, self.data2, start, count,
                    self.epochs_per_subj, Epad, out=self._zbuf[bidx],
                    raw=self._raw_split)
                ev = torch.cuda.Event()
                ev.record(corr_stream)
            if pending is not None:
                _consume(*pending)
            pending = (z, ev, bidx)
        _consume(*pending)

        cur = torch.cuda.current_stream(self.device)
        cur.wait_stream(gram_stream)
        cur.wait_stream(corr_stream)
        for g in grams:   # allocated on gram_stream, consumed on cur
            g.record_stream(cur)
        gram = torch.cat(grams, dim=0)
        if shrink:
            _shrink_(gram)
        return gram

    def _duo_pipeline(self, chunks, shrink=True, consumer=None):
        """Single-stream duo launches: each kernel carries the raw-corr
        blocks of chunk i AND the Gram(+normalize) blocks of chunk i-1
        in one grid — HIP streams do not co-schedule the two kernels
        (measured exactly serial), a shared grid forces CU-level
        co-residency of the VALU-bound and latency-bound waves.

        consumer, if given, is called as consumer(g, start, count)
        right after each chunk's [count, E, E] Gram is enqueued (in
        chunk order, stream-ordered on the current stream) and the
        grams are NOT accumulated — the pipeline returns None.  This
        is how the voxel selector overlaps the per-chunk SVM CV with
        the remaining duo sweep.  Grams arrive already shrunk when
        shrink=True (the partial-sum + shrink of chunk i-2 rides the
        duo grid of launch i as a third block population)."""
        ext = ops.load_extension()
        E = self.num_epochs
        Epad = ((E + 63) // 64) * 64
        VB = self.data2.shape[2]
        P = self.epochs_per_subj
        max_count = max(c for _, c in chunks)
        if getattr(self, "_zbuf", None) is None or \
                self._zbuf[0].shape[0] < max_count or \
                self._zbuf[0].shape[1] != Epad or \
                self._zbuf[0].dtype != torch.bfloat16:
            self._zbuf = [
                torch.zeros((max_count, Epad, VB),
                            dtype=torch.bfloat16, device=self.device)
                for _ in range(2)]
        nsplit = int(os.environ.get("BRAINIAK_GRAM_NSPLIT", "0")) or \
            max(1, (8191 + max_count) // max_count)
        nsplit = min(nsplit, (VB + 127) // 128)

        grams = []

        def emit(g, start_v, count_v):
            if consumer is not None:
                consumer(g, start_v, count_v)
            else:
                grams.append(g)

        def slice_epochs(g):
            return g if Epad == E else g[:, :E, :E].contiguous()

        prev = None                     # (z, count)
        pending = []                    # [(gp, count, start)] FIFO
        for i, (start, count) in enumerate(chunks):
            z_i = self._zbuf[i % 2]
            if prev is None:
                ext.fcma_corr_norm_z(self.data, self.data2, start,
                                     count, P, Epad, out=z_i, raw=True)
            else:
                zp, cp = prev
                gp = torch.empty((nsplit, cp, Epad, Epad),
                                 dtype=torch.float32,
                                 device=self.device)
                gout = None
                extra = {}
                if pending and os.environ.get("BRAINIAK_NO_GSUM"):
                    # A/B fallback: torch partial-sum + shrink
                    pgp, pcnt, pstart = pending.pop(0)
                    g = pgp.sum(0)
                    if shrink:
                        _shrink_(g)
                    emit(slice_epochs(g), pstart, pcnt)
                elif pending:
                    # chunk i-2's partials are complete: their
                    # reduction + magnitude shrink join this launch's
                    # grid as a third block population
                    pgp, pcnt, pstart = pending.pop(0)
                    gout = torch.empty((pcnt, Epad, Epad),
                                       dtype=torch.float32,
                                       device=self.device)
                    extra = dict(Gsum_part=pgp, Gsum_out=gout,
                                 shrink=shrink)
                ext.fcma_corr_gram_duo(self.data, self.data2, start,
                                       count, P, z_i.narrow(0, 0, count),
                                       Zprev=zp, Gpart=gp, **extra)
                if gout is not None:
                    emit(slice_epochs(gout), pstart, pcnt)
                pending.append((gp, cp, chunks[i - 1][0]))
            prev = (z_i.narrow(0, 0, count), count)
        # tail: at most one chunk's partials left, plus the final
        # chunk's gram (no following launch to ride)
        for pgp, pcnt, pstart in pending:
            g = pgp.sum(0)
            if shrink:
                _shrink_(g)
            emit(slice_epochs(g), pstart, pcnt)
        zp, cp = prev
        g = ops.fcma_gram_bf16(zp, norm_P=P)
        if shrink:
            _shrink_(g)
        emit(slice_epochs(g), chunks[-1][0], cp)
        if consumer is not None:
            return None
        return torch.cat(grams, dim=0)
