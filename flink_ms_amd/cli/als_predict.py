"""Interactive ALS prediction client (reference ALSPredict.java).

Usage: ``als_predict <jobID> [<jobManagerHost>] [<jobManagerPort>]`` — jobID
is accepted for CLI parity (the HTTP server needs no job id).  Reads
``user,item`` lines, queries "<u>-U" and "<i>-I" from ALS_MODEL and prints
the dot product (ALSPredict.java:60-86).
"""
import sys

from ..serving.client import QueryClientHelper


def main(argv=None) -> int:
    args = sys.argv[1:] if argv is None else argv
    if not args:
        print("Missing required job ID argument. "
              "Usage: ./ALSPredict <jobID> [jobManagerHost] [jobManagerPort]")
        return 1
    host = args[1] if len(args) > 1 else "localhost"
    port = int(args[2]) if len(args) > 2 else 6123
    print(f"Using JobManager {host}:{port}")
    print("Enter <User,Item> to predict.")
    with QueryClientHelper(host, port) as client:
        for line in sys.stdin:
            key = line.upper().strip()
            if not key:
                continue
            print(f"[info] Querying the model for <user,item> pair '{key}'")
            try:
                user_id, item_id = key.split(",")
                user = client.query_state("ALS_MODEL", f"{user_id}-U")
                item = client.query_state("ALS_MODEL", f"{item_id}-I")
                if user is not None and item is not None:
                    uf = [float(x) for x in user[1].split(";")]
                    vf = [float(x) for x in item[1].split(";")]
                    prediction = sum(a * b for a, b in zip(uf, vf))
                    print(f"ALS Prediction =  {prediction:f} ")
                else:
                    print("User or Item Factors do not exist in the model "
                          f"for the query: {key}")
            except Exception as e:  # noqa: BLE001
                print("Query failed because of the following Exception:")
                print(e)
    return 0


if __name__ == "__main__":
    sys.exit(main())
