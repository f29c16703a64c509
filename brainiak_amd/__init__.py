"""brainiak_amd — MI355X-native fMRI analysis toolkit.

A brand-new AMD-native framework with the capability set of BrainIAK
(see /root/reference): FCMA, SRM/DetSRM/RSRM/SS-SRM/FastSRM,
distributed Searchlight, ISC/ISFC + nonparametric stats, event
segmentation, TFA/HTFA, BRSA/GBRSA, matrix-normal models, IEM, HPO,
and fmrisim — behind the same sklearn-style fit/transform API.

Design (MI355X-first):
 - PyTorch-ROCm tensors are the compute lingua franca; every estimator
   also accepts/returns numpy for API parity with the reference.
 - Hot kernels (FCMA correlation+normalization+SVM-kernel pipeline,
   SRM Procrustes) are hand-written HIP/CDNA4 (gfx950) kernels in
   ``brainiak_amd.ops`` — MFMA tiles, LDS staging, 64-wide wavefronts.
 - Distribution is one process per GPU with ``torch.distributed``
   (RCCL over the xGMI mesh; gloo on CPU) — see ``brainiak_amd.parallel``.
   No MPI, no CUDA-compat shims, no Triton.
"""

__version__ = "0.1.0"

__all__ = [
    "eventseg",
    "factoranalysis",
    "fcma",
    "funcalign",
    "hyperparamopt",
    "image",
    "io",
    "isc",
    "matnormal",
    "nifti",
    "ops",
    "parallel",
    "reconstruct",
    "reprsimil",
    "searchlight",
    "utils",
]
