"""Correlation-based voxel selection (FCMA stage 1).

API parity with the reference ``VoxelSelector``
(ref src/brainiak/fcma/voxelselector.py:56-516): same constructor
arguments, same ``run(clf)`` → sorted [(voxel_id, score)] contract, same
numerics for the correlation → Fisher-z → Gram → CV chain.

MI355X-first redesign: the reference's MPI master-worker task farm
(voxelselector.py:145-282) — dynamic 64-voxel tasks, pickled messages,
CPU OpenMP/BLAS workers — becomes *static voxel sharding*: each of the N
ranks owns a contiguous slice of the selected-voxel range, streams it
through the GPU pipeline (brainiak_amd.fcma.core, HIP kernels on
gfx950) in ``voxel_unit`` chunks, and one RCCL all-gather at the end
collects the (voxel, score) pairs.  Intra-node xGMI makes farm-style
latency hiding pointless; removing the master frees a GPU.
"""

import logging
import os
from typing import Optional

import numpy as np
import torch

from ..parallel import DistContext
from ..utils.timing import stage_timer
from .core import CorrelationPipeline
from .svm import cross_validate_voxels

logger = logging.getLogger(__name__)

__all__ = ["VoxelSelector"]


class VoxelSelector:
    """Distributed correlation-based voxel selection.

    Parameters (reference-compatible)
    ---------------------------------
    labels : list of per-epoch condition labels (len == num_epochs).
    epochs_per_subj : epochs per subject (all subjects equal).
    num_folds : CV folds.
    raw_data : list of [epoch_len, nVoxels] z-scored epoch matrices.
    raw_data2 : optional second-mask epochs (cross-correlation FCMA).
    voxel_unit : chunk size streamed through the pipeline per step.
    process_num : CPU pool size for the sklearn CV path (0 = in-process).
    master_rank : accepted for API parity; unused (no master here).

    New (MI355X) parameters
    -----------------------
    comm : DistContext (defaults to the process-global one).
    device : torch device override.
    use_gpu_svm : solve the per-voxel SVM duals batched on device
        (default when the pipeline runs on GPU).
    """

    def __init__(self, labels, epochs_per_subj, num_folds, raw_data,
                 raw_data2=None, voxel_unit=None, process_num=4,
                 master_rank=0, comm=None, device=None,
                 use_gpu_svm: Optional[bool] = None):
        self.labels = np.asarray(labels)
        self.epochs_per_subj = epochs_per_subj
        self.num_folds = num_folds
        self.raw_data = raw_data
        self.raw_data2 = raw_data2
        if voxel_unit is None:
            # GPU chunks amortize kernel launches; 64 is the reference's
            # CPU task size (voxelselector.py:89)
            voxel_unit = 1024 if torch.cuda.is_available() else 64
        self.num_voxels = raw_data[0].shape[1]
        self.num_voxels2 = (raw_data2[0].shape[1] if raw_data2 is not None
                            else self.num_voxels)
        self.voxel_unit = voxel_unit
        self.process_num = process_num
        self.master_rank = master_rank
        self.ctx = comm if isinstance(comm, DistContext) else DistContext()
        self.device = device
        self.use_gpu_svm = use_gpu_svm
        if raw_data2 is not None and len(raw_data) != len(raw_data2):
            raise ValueError('The raw data lists must have the same number '
                             'of elements for computing the correlations '
                             'element by element')
        if self.num_voxels == 0 or self.num_voxels2 == 0:
            raise ValueError('Zero processed voxels')
        if len(self.labels) != len(raw_data):
            raise ValueError('Number of labels must equal number of epochs')

    def _clf_params(self, clf):
        """Extract (is_precomputed_svm, C, tol) from an sklearn-style clf."""
        try:
            import sklearn.svm
            if isinstance(clf, sklearn.svm.SVC) and \
                    clf.kernel == 'precomputed':
                return True, float(clf.C), float(getattr(clf, 'tol', 1e-3))
        except ImportError:  # pragma: no cover
            pass
        return False, 1.0, 1e-3

    def run(self, clf):
        """Score every voxel; returns [(voxel_id, score)] sorted by score
        descending (ties by voxel id, matching the reference's sort)."""
        pipeline = CorrelationPipeline(
            self.raw_data, self.raw_data2, self.epochs_per_subj,
            device=self.device)
        precomputed, C, tol = self._clf_params(clf)
        use_gpu_svm = self.use_gpu_svm
        if use_gpu_svm is None:
            use_gpu_svm = pipeline.device.type == "cuda"

        my = self.ctx.shard(self.num_voxels)
        results = []
        if precomputed and pipeline.device.type == "cuda":
            # whole-shard path: stream-pipelined corr+norm/Gram kernels,
            # then ONE batched CV launch over every scored voxel
            chunks = []
            start = my.start
            while start < my.stop:
                count = min(self.voxel_unit, my.stop - start)
                chunks.append((start, count))
                start += count
            scores = None
            if use_gpu_svm:
                with stage_timer("voxel selection (duo sweep + "
                                 "overlapped CV)", logger,
                                 sync_device=pipeline.device):
                    scores = self._overlapped_cv(pipeline, chunks, C,
                                                 tol)
            if scores is None:
                with stage_timer("correlation/Gram pipeline", logger,
                                 sync_device=pipeline.device):
                    kernels = pipeline.pipelined_kernel_matrices(chunks)
                with stage_timer("cross validation", logger,
                                 sync_device=pipeline.device):
                    if use_gpu_svm:
                        scores = cross_validate_voxels(
                            kernels, self.labels, self.num_folds, C=C,
                            tol=tol)
                    else:
                        scores = self._sklearn_cv(
                            clf, kernels.cpu().numpy())
            results.extend((my.start + i, float(scores[i]))
                           for i in range(my.stop - my.start))
        else:
            start = my.start
            while start < my.stop:
                count = min(self.voxel_unit, my.stop - start)
                scores = self._score_chunk(pipeline, clf, start, count,
                                           precomputed, C, tol,
                                           use_gpu_svm)
                results.extend(
                    (start + i, float(scores[i])) for i in range(count))
                start += count
        logger.info('rank %d scored voxels [%d, %d)', self.ctx.rank,
                    my.start, my.stop)

        if self.ctx.is_distributed:
            gathered = self.ctx.all_gather_object(results)
            results = [r for part in gathered for r in part]
        results.sort(key=lambda t: (-t[1], t[0]))
        return results

    def _overlapped_cv(self, pipeline, chunks, C, tol):
        """Duo sweep with the per-chunk SVM CV enqueued on a SIDE
        stream: the SMO grid for chunk i (512 x folds wavefronts, tiny
        next to the duo grid) fills CU slots as the duo kernel for
        chunk i+1 drains, instead of paying the whole CV serially
        after the sweep (~6 % of a whole-brain pass).  Accuracy stays
        on device until one download at the end.  Returns None when
        the duo path or the fold sizes do not apply, and the caller
        falls back to the batched post-pass CV."""
        if (not pipeline._hip_path or not pipeline._raw_split
                or len(chunks) <= 1
                or os.environ.get("BRAINIAK_NO_DUO")
                or os.environ.get("BRAINIAK_NO_CV_OVERLAP")):
            return None
        from .svm import FoldPlan, svm_cv_device
        try:
            plan = FoldPlan(self.labels, self.num_folds,
                            pipeline.device)
        except ValueError:
            return None
        if plan.max_n > 128:
            return None
        dev = pipeline.device
        prio = int(os.environ.get("BRAINIAK_CV_PRIO", "0"))
        cv_stream = torch.cuda.Stream(device=dev, priority=prio)
        accs = []

        def consume(g, start, count):
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream(dev))
            with torch.cuda.stream(cv_stream):
                cv_stream.wait_event(ev)
                accs.append(svm_cv_device(g, plan, C, tol))
                # g was allocated on the main stream; keep the
                # allocator from recycling it under the CV reads
                g.record_stream(cv_stream)

        pipeline._duo_pipeline(chunks, consumer=consume)
        torch.cuda.current_stream(dev).wait_stream(cv_stream)
        return torch.cat(accs).cpu().numpy()

    def _score_chunk(self, pipeline, clf, start, count, precomputed, C,
                     tol, use_gpu_svm):
        if precomputed:
            kernels = pipeline.chunk_kernel_matrices(start, count)
            if use_gpu_svm and kernels.is_cuda:
                return cross_validate_voxels(kernels, self.labels,
                                             self.num_folds, C=C, tol=tol)
            return self._sklearn_cv(clf, kernels.cpu().numpy())
        # non-precomputed classifier: CV directly on the normalized
        # correlation vectors (reference behaviour for e.g. logistic clf)
        corr = pipeline.correlate_chunk(start, count)
        from .core import normalize_correlation_
        normalize_correlation_(corr, self.epochs_per_subj)
        data = corr.cpu().numpy()
        return self._sklearn_cv(clf, data)

    def _sklearn_cv(self, clf, data):
        from sklearn import base, model_selection
        skf = model_selection.StratifiedKFold(n_splits=self.num_folds,
                                              shuffle=False)
        scores = np.empty(data.shape[0])
        for i in range(data.shape[0]):
            cv = model_selection.cross_val_score(
                base.clone(clf), data[i], y=self.labels, cv=skf, n_jobs=1)
            scores[i] = cv.mean()
        return scores
