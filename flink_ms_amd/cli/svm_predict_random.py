"""SVM load generator (reference SVMPredictRandom.java).

Flags: --maxNoOfFeatures (required), --numQueries (1000),
--minPercentageOfFeatures (10), --outputDecisionFunction (false),
--thresholdValue (0.0), --jobManagerHost/Port, --queryTimeout,
--outputFile.  Output CSV: ``queryId,nnz,prediction,millis``.
"""
import sys

from ..serving.client import QueryClientHelper
from ..serving.loadgen import svm_predict_random
from ..utils.params import Params


def main(argv=None) -> int:
    p = Params.from_args(sys.argv[1:] if argv is None else argv)
    client = QueryClientHelper(p.get("jobManagerHost", "localhost"),
                               p.get_int("jobManagerPort", 6123),
                               p.get_int("queryTimeout", 5))
    res = svm_predict_random(
        max_no_of_features=p.get_required_int("maxNoOfFeatures"),
        num_queries=p.get_int("numQueries", 1000),
        min_percentage_of_features=p.get_int("minPercentageOfFeatures", 10),
        output_decision_function=p.get_bool("outputDecisionFunction", False),
        threshold_value=p.get_float("thresholdValue", 0.0),
        client=client)
    client.close()
    print("Output is written in the format: queryID, nnz, prediction, "
          "timeInMillis")
    if p.has("outputFile"):
        res.write_csv(p.get("outputFile"), "queryId,nnz,prediction,millis")
    else:
        for row in res.csv_rows:
            print(row)
    print(res.summary())
    return 0


if __name__ == "__main__":
    sys.exit(main())
