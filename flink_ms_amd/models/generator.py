"""Synthetic model generators (model-generator module rebuild).

Reference: model-generator/src/main/scala/de/tub/it4bi/ALSModelGenerator.scala
and SVMModelGenerator.scala — fabricate arbitrarily large models in the text
formats "only for testing the latency and throughput. Not for quality".

Distribution shapes preserved:
- ALS factors: ``nextDouble()/nextDouble() * latentFactors`` — a heavy-tailed
  ratio distribution (ALSModelGenerator.scala:28-32); ids start at 1.
- SVM weights: ~50% exact zeros (emitted as the bare token "0", matching the
  reference's Int-typed zero), otherwise ~uniform in [-10, 10] via binary
  subdivision (SVMModelGenerator.scala:26-52); keys are 0-based
  ``bucket*range .. bucket*range+range-1``.
"""

from __future__ import annotations

from typing import Iterator

import torch

from ..utils.textio import als_factor_row, java_double_to_string


def generate_als_model(num_users: int, num_items: int, latent_factors: int,
                       seed: int = 42) -> Iterator[str]:
    g = torch.Generator(device="cpu").manual_seed(seed)

    def factors() -> list:
        a = torch.rand(latent_factors, generator=g, dtype=torch.float64)
        b = torch.rand(latent_factors, generator=g, dtype=torch.float64)
        return (a / b * latent_factors).tolist()

    for u in range(1, num_users + 1):
        yield als_factor_row(u, "U", factors())
    for i in range(1, num_items + 1):
        yield als_factor_row(i, "I", factors())


def generate_svm_model(num_features: int, range_size: int,
                       seed: int = 42) -> Iterator[str]:
    """Range-partitioned random weight rows ``<bucket>,<i>:<w>;...``."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    num_buckets = (num_features + range_size - 1) // range_size
    for bucket in range(num_buckets):
        start = bucket * range_size
        zeros = torch.rand(range_size, generator=g) < 0.5
        vals = torch.rand(range_size, generator=g, dtype=torch.float64) * 20 - 10
        parts = []
        for j in range(range_size):
            key = start + j
            if bool(zeros[j]):
                parts.append(f"{key}:0")  # Int-typed zero, as the reference
            else:
                parts.append(f"{key}:{java_double_to_string(float(vals[j]))}")
        yield f"{bucket}," + ";".join(parts)
