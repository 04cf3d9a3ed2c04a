"""ALS load generator (reference ALSPredictRandom.java).

Flags (ALSPredictRandom.java:30-46): --jobId (parity), --jobManagerHost,
--jobManagerPort, --queryTimeout (5), --numQueries (1000), --lowerUserId /
--upperUserId / --lowerItemId / --upperItemId, --outputFile.
Output CSV: ``uId,iId,prediction,millis``.
"""
import sys

from ..serving.client import QueryClientHelper
from ..serving.loadgen import als_predict_random
from ..utils.params import Params

INT_MAX = 2 ** 31 - 1


def main(argv=None) -> int:
    p = Params.from_args(sys.argv[1:] if argv is None else argv)
    client = QueryClientHelper(p.get("jobManagerHost", "localhost"),
                               p.get_int("jobManagerPort", 6123),
                               p.get_int("queryTimeout", 5))
    res = als_predict_random(
        num_queries=p.get_int("numQueries", 1000),
        lower_user_id=p.get_int("lowerUserId", 0),
        upper_user_id=p.get_int("upperUserId", INT_MAX),
        lower_item_id=p.get_int("lowerItemId", 0),
        upper_item_id=p.get_int("upperItemId", INT_MAX),
        client=client)
    client.close()
    print("Output is written in the format: userID, itemID, prediction, "
          "timeInMillis")
    if p.has("outputFile"):
        res.write_csv(p.get("outputFile"), "uId,iId,prediction,millis")
    else:
        for row in res.csv_rows:
            print(row)
    print(res.summary())
    return 0


if __name__ == "__main__":
    sys.exit(main())
