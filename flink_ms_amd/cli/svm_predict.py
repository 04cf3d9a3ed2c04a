"""Interactive SVM classification client (reference SVMPredict.java).

Usage: ``svm_predict <jobID> [host] [port] [outputDecisionFunction]
[threshold]``.  Reads sparse ``id:val id:val`` lines, one state lookup per
feature, margin or +-1 output (SVMPredict.java:55-95).
"""
import sys

from ..serving.client import QueryClientHelper


def main(argv=None) -> int:
    args = sys.argv[1:] if argv is None else argv
    if not args:
        print("Missing required job ID argument. "
              "Usage: ./SVMPredict <jobID> [host] [port] "
              "[outputDecisionFunction] [threshold]")
        return 1
    host = args[1] if len(args) > 1 else "localhost"
    port = int(args[2]) if len(args) > 2 else 6123
    output_decision = len(args) > 3 and args[3].lower() == "true"
    threshold = float(args[4]) if len(args) > 4 else 0.0
    print(f"Using JobManager {host}:{port}")
    print("Enter Vector data to predict.")
    with QueryClientHelper(host, port) as client:
        for line in sys.stdin:
            if not line.strip():
                continue
            print(f"[info] Querying the model for vector '{line.strip()}' ")
            try:
                raw = 0.0
                for tok in line.strip().split(" "):
                    fid, val = tok.split(":")
                    hit = client.query_state("SVM_MODEL", fid)
                    if hit is not None:
                        raw += float(hit[1]) * float(val)
                    else:
                        print(f"Could not find the value for feature ID: {fid} ")
                if output_decision:
                    prediction = raw
                else:
                    prediction = 1.0 if raw > threshold else -1.0
                print(f"SVM Prediction =  {prediction:f} ")
            except Exception as e:  # noqa: BLE001
                print("Query failed because of the following Exception:")
                print(e)
    return 0


if __name__ == "__main__":
    sys.exit(main())
