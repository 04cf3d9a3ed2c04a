"""Synthetic SVM model generator (reference model-generator/.../SVMModelGenerator.scala).

Flags: --numFeatures --range (required), --parallelism (2), --output.
"""
import sys

from ..models.generator import generate_svm_model
from ..utils.params import Params


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    rows = generate_svm_model(
        params.get_required_int("numFeatures"),
        params.get_required_int("range"),
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
