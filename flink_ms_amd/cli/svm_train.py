"""SVM training job (reference flink-svm/.../SVMImpl.scala rebuild).

Flags (SVMImpl.scala:19-48): --training (required), --blocks (10),
--iteration (10), --range (1000), --partition (bool), --output.
"""
import sys

from ..data.libsvm import read_libsvm
from ..models.svm import SVMConfig, SVMTrainer
from ..parallel.dist import init_from_env
from ..utils.params import Params


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    ctx = init_from_env()
    csr, labels, _ = read_libsvm(params.get_required("training"))
    # flink-ml uses 0/1-coded labels from readLibSVM as-is; normalize to +-1
    y = labels.sign() + (labels == 0).float() * -1.0
    trainer = SVMTrainer(SVMConfig(
        blocks=params.get_int("blocks", 10),
        iterations=params.get_int("iteration", 10),
    ), ctx)
    trainer.setup(csr, y)
    model = trainer.fit()

    out = params.get("output")
    range_size = params.get_int("range", 1000)
    import io
    buf = io.StringIO()
    if params.get_bool("partition", False):
        model.write_range_partitioned(buf, range_size)
    else:
        model.write_flat(buf)
    if out:
        with open(out, "w") as f:
            f.write(buf.getvalue())
        print("[SVM] model-fitting done")
    else:
        print("Printing result to stdout. Use --output to specify output path.")
        print(buf.getvalue(), end="")
    return 0


if __name__ == "__main__":
    sys.exit(main())
