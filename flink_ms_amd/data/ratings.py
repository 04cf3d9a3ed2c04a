"""Rating-triple input layer.

The reference trains ALS from CSV ``(Int, Int, Double)`` triples with a comma
or tab delimiter and optional header skip (reference
flink-als/.../ALSImpl.scala:22-32); SGD/MSE default to tab (SGD.java:106,
MSE.java:42).  This module provides the same loaders plus the synthetic
generators the BASELINE configs need (no network: datasets are fabricated at
the published shapes with random values).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Tuple

import torch


@dataclass(frozen=True)
class RatingsShape:
    num_users: int
    num_items: int
    num_ratings: int


# MovieLens shapes (counts only — data itself is synthetic, random-init).
ML100K_SHAPE = RatingsShape(943, 1682, 100_000)
ML25M_SHAPE = RatingsShape(162_541, 59_047, 25_000_095)


def synthetic_ratings(
    shape: RatingsShape,
    seed: int = 42,
    device: str = "cpu",
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Uniform-random (user, item, rating) triples of the given shape.

    Ratings are uniform in [0.5, 5.0] (MovieLens value range).  Duplicate
    (user, item) pairs are possible as in any sampled benchmark load; ALS's
    normal equations are well-defined regardless.
    """
    g = torch.Generator(device="cpu").manual_seed(seed)
    users = torch.randint(0, shape.num_users, (shape.num_ratings,), generator=g, dtype=torch.int32)
    items = torch.randint(0, shape.num_items, (shape.num_ratings,), generator=g, dtype=torch.int32)
    vals = torch.rand(shape.num_ratings, generator=g, dtype=torch.float32) * 4.5 + 0.5
    if device != "cpu":
        users, items, vals = users.to(device), items.to(device), vals.to(device)
    return users, items, vals


def load_ratings_csv(
    path: str,
    field_delimiter: str = "comma",
    ignore_first_line: bool = True,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Read ``(Int, Int, Double)`` rating triples.

    ``field_delimiter`` is the reference's flag vocabulary: "comma" or "tab"
    (ALSImpl.scala:22-27); a literal one-character delimiter is also accepted.
    """
    delim = {"comma": ",", "tab": "\t"}.get(field_delimiter, field_delimiter)
    users, items, vals = [], [], []
    with open(path) as f:
        first = True
        for line in f:
            if first and ignore_first_line:
                first = False
                continue
            first = False
            line = line.strip()
            if not line:
                continue
            u, i, r = line.split(delim)[:3]
            users.append(int(u))
            items.append(int(i))
            vals.append(float(r))
    return (
        torch.tensor(users, dtype=torch.int32),
        torch.tensor(items, dtype=torch.int32),
        torch.tensor(vals, dtype=torch.float32),
    )
