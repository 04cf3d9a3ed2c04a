"""Mean-vector job (reference flink-als/.../ALSMeanVector.scala rebuild).

Flags: --type item|user (required), --input (file OR a directory of part
files, as distributed training writes), --output.
"""
import sys

from ..models.mean_vector import mean_vector_rows
from ..serving.app import _read_rows
from ..utils.params import Params


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    t = params.get_required("type")
    if t == "item":
        factor_type = "I"
    elif t == "user":
        factor_type = "U"
    else:
        raise ValueError("specify type as either 'item' or 'user'.")
    row = mean_vector_rows(_read_rows(params.get_required("input")),
                           factor_type)
    if params.has("output"):
        with open(params.get("output"), "w") as f:
            f.write(row + "\n")
    else:
        print("Printing results to stdout. Use --output to specify output location")
        print(row)
    return 0


if __name__ == "__main__":
    sys.exit(main())
