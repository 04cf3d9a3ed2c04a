"""Model publisher (reference ALSKafkaProducer.java / SVMKafkaProducer.java).

Streams model text rows from --input (file or directory of part files,
nested files included, ALSKafkaProducer.java:24-26) into the serving
server's ingest endpoint (the Kafka-topic replacement).

Flags: --input (required), --model als|svm (the "topic"), --server host,
--port, --batchSize.
"""
import sys

from ..serving.app import _read_rows
from ..serving.client import QueryClientHelper
from ..utils.params import Params


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    rows = _read_rows(params.get_required("input"))
    model = params.get("model", params.get("topic", "als"))
    if model not in ("als", "svm"):
        model = "als" if "als" in model.lower() else "svm"
    batch = params.get_int("batchSize", 10000)
    n = 0
    with QueryClientHelper(params.get("server", "localhost"),
                           params.get_int("port", 6123)) as client:
        for s in range(0, len(rows), batch):
            n += client.ingest_rows(model, rows[s:s + batch])
    print(f"published {n} rows to {model} model store")
    return 0


if __name__ == "__main__":
    sys.exit(main())
