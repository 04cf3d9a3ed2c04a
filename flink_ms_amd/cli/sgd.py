"""Online SGD job (reference als-ms/.../qs/SGD.java, v1 semantics).

Streams rating rows from --input (file; --mode continuous re-polls for
appended rows every --interval ms, mode once reads it once: SGD.java:48-64),
queries the serving store for the current factors with MEAN-U / MEAN-I
cold-start fallback fetched once at open (SGD.java:127-152), applies the
v1 "simultaneous" update (both vectors from the OLD copies) and writes the
updated rows back.

--outputMode kafka sends the rows to the serving server's ingest endpoint
(closing the loop the reference closes through the Kafka topic + consumer);
--outputMode hdfs appends them to --outputPath instead.

Flags (SGD.java:44-166): --input, --mode continuous|once (required),
--interval (60000), --fieldDelimiter (tab), --outputMode (kafka),
--topic/--outputPath, --jobId (parity), --jobManagerHost (localhost),
--jobManagerPort (6123), --queryTimeout (5), --learningRate (0.1),
--userRegularization (0.0), --itemRegularization (0.0), --userMean,
--itemMean.
"""
import sys
import time

from ..serving.client import QueryClientHelper
from ..utils.params import Params


def run_once(client: QueryClientHelper, rows, params, out_file=None) -> int:
    kw = dict(
        field_delimiter=params.get("fieldDelimiter", "\t"),
        learning_rate=params.get_float("learningRate", 0.1),
        user_regularization=params.get_float("userRegularization", 0.0),
        item_regularization=params.get_float("itemRegularization", 0.0),
        user_mean=params.get("userMean"),
        item_mean=params.get("itemMean"),
        v0_semantics=params.get_bool("v0", False),  # SGDV0.java variant
    )
    if not rows:
        return 0
    # --gpuBatch: route through the K4 batched device kernel
    # (/sgd/update_batch) instead of the scalar fp64 loop — v1 semantics
    # at the store's bf16 precision, MEAN cold-start preserved
    if params.get_bool("gpuBatch", False):
        def _send(rs):
            return client.sgd_update_batch(
                rs, field_delimiter=kw["field_delimiter"],
                learning_rate=kw["learning_rate"],
                user_regularization=kw["user_regularization"],
                item_regularization=kw["item_regularization"])
    else:
        def _send(rs):
            return client.sgd_update(rs, **kw)
    if params.get("outputMode", "kafka") == "hdfs":
        # compute on the server but persist rows to the output path
        resp = _send(rows)
        with open(params.get_required("outputPath"), "a") as f:
            for row in resp["rows"]:
                f.write(row + "\n")
    else:
        resp = _send(rows)
    for rid in resp.get("nan_records", []):
        print(f"NaN detected for: {rid}")
    return resp["updated"]


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    mode = params.get_required("mode")
    interval_ms = params.get_int("interval", 60000)
    path = params.get_required("input")
    client = QueryClientHelper(params.get("jobManagerHost", "localhost"),
                               params.get_int("jobManagerPort", 6123),
                               params.get_int("queryTimeout", 5))
    # MEAN fallback must exist up front (SGD.java:149-152)
    if (client.query_state("ALS_MODEL", "MEAN-U") is None
            and params.get("userMean") is None) or (
            client.query_state("ALS_MODEL", "MEAN-I") is None
            and params.get("itemMean") is None):
        raise RuntimeError("Unable to load User mean or item mean factors.")
    seen = 0
    total = 0
    polls = 0
    # --maxPolls: operational bound for tests/drains (not a reference flag;
    # the reference job streams forever)
    max_polls = params.get_int("maxPolls", 0)
    while True:
        with open(path) as f:
            rows = [line for line in f.read().splitlines() if line.strip()]
        new_rows = rows[seen:]
        seen = len(rows)
        total += run_once(client, new_rows, params)
        polls += 1
        if mode != "continuous" or (max_polls and polls >= max_polls):
            break
        time.sleep(interval_ms / 1000.0)
    print(f"applied {total} online updates")
    client.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
