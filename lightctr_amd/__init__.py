"""lightctr_amd — MI355X-native sparse CTR training framework.

A from-scratch rebuild of the capabilities of cnkuangshi/LightCTR
(reference surveyed in SURVEY.md) designed for AMD Instinct MI355X
(gfx950 / CDNA4): hand-written HIP kernels for the hot sparse/dense ops,
PyTorch-ROCm as the tensor substrate, RCCL over xGMI for all
cross-GPU communication (ring all-reduce data parallelism and a
hash-sharded parameter-server mode).

Layout:
  lightctr_amd.ops       — compute ops: HIP kernels (GPU) + torch reference (CPU oracle)
  lightctr_amd.models    — model zoo: FM, FFM, NFM, Wide&Deep, GBM, GMM, PLSA,
                           embeddings, VAE, CNN, RNN(LSTM+attention)
  lightctr_amd.parallel  — RCCL ring DP + sharded PS (all-to-all) paths
  lightctr_amd.data      — libffm loader, synthetic Criteo-shaped generator
  lightctr_amd.utils     — metrics (AUC), updaters, checkpointing, watchdog
  lightctr_amd.engine    — training drivers / DAG-style pipeline scheduler
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
