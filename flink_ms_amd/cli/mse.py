"""MSE evaluation job (reference als-ms/.../evaluation/MSE.java).

Flags (MSE.java:40-105): --input, --fieldDelimiter (tab), --jobId (parity),
--jobManagerHost, --jobManagerPort, --queryTimeout, --output.
Scores the rating file against the served model and prints/writes the MSE.
"""
import sys

import httpx

from ..utils.params import Params


def main(argv=None) -> int:
    params = Params.from_args(sys.argv[1:] if argv is None else argv)
    host = params.get("jobManagerHost", "localhost")
    port = params.get_int("jobManagerPort", 6123)
    delim = params.get("fieldDelimiter", "\t")
    with open(params.get_required("input")) as f:
        rows = [line for line in f.read().splitlines() if line.strip()]
    r = httpx.post(f"http://{host}:{port}/mse",
                   json={"ratings": rows, "field_delimiter": delim},
                   timeout=float(params.get_int("queryTimeout", 5) * 60))
    r.raise_for_status()
    result = r.json()
    if params.has("output"):
        with open(params.get("output"), "w") as f:
            f.write(f"{result['mse']}\n")
    print(f"MSE = {result['mse']} (scored {result['scored']}, "
          f"skipped {result['skipped']})")
    return 0


if __name__ == "__main__":
    sys.exit(main())
