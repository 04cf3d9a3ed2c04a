"""flink_ms_amd — MI355X-native ALS recommender + CoCoA-SVM training/serving stack.

A from-scratch rebuild of the capabilities of mmziyad/flink-ms (Apache Flink
1.3.1 train-then-serve ML stack) designed MI355X-first:

- PyTorch-ROCm driver loops replace the Flink DataSet bulk-iteration engine.
- Hot ops (ALS Gramian assembly + batched Cholesky solve, CoCoA-SVM local SDCA,
  serving dot/SGD updates) are hand-written CDNA4 HIP kernels (MFMA tiles,
  LDS-staged factor blocks) in `flink_ms_amd/ops/csrc/`.
- The per-iteration user<->item factor shuffle and the SVM primal aggregate run
  as RCCL collectives over xGMI (`flink_ms_amd/parallel/`).
- The flink-queryable-client query surface (reference
  flink-queryable-client/.../ALSPredict.java etc.) is re-exposed over REST with
  byte-identical text payloads (`flink_ms_amd/serving/`).

Layout:
  utils/     text model-row codecs (Java payload parity), logging, params
  data/      synthetic generators + CSV/LibSVM loaders + blocked sparse layouts
  ops/       HIP kernel wrappers + pure-torch reference implementations
  parallel/  torch.distributed (RCCL/gloo) process-group + factor exchange
  models/    ALS / CoCoA-SVM trainers, online SGD, MSE, mean-vector, generators
  serving/   GPU-resident keyed model store + REST server + clients/loadgen
  cli/       job entry points mirroring the reference CLI flag surface
"""

__version__ = "0.1.0"

# _hip_ops links libtorch: importing it before torch's python runtime is
# initialized segfaults in tensor allocation.  Importing torch here makes
# `from flink_ms_amd import _hip_ops` safe in any order (observed with the
# standalone KvServer / parser probes).
import torch  # noqa: E402,F401
