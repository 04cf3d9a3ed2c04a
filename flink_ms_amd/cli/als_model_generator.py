"""Synthetic ALS model generator (reference model-generator/.../ALSModelGenerator.scala).

Flags: --numUsers --numItems --latentFactors (required), --parallelism (2),
--output.  "Only for testing the latency and throughput. Not for quality."
"""
import sys

from ..models.generator import generate_als_model
from ..utils.params import Params


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    rows = generate_als_model(
        params.get_required_int("numUsers"),
        params.get_required_int("numItems"),
        params.get_required_int("latentFactors"),
    )
    if params.has("output"):
        with open(params.get("output"), "w") as f:
            for row in rows:
                f.write(row + "\n")
    else:
        for row in rows:
            print(row)
    return 0


if __name__ == "__main__":
    sys.exit(main())
