"""Text model-row codecs with Java payload parity.

The reference's durable model format is text rows; every
producer formats doubles with Java's ``Double.toString`` and every consumer
re-parses them.  Format contracts reproduced here:

- ALS factor row      ``"<id>,<U|I>,<f1>;<f2>;...;<fk>"``
  (reference flink-als/.../ALSImpl.scala:83-85, ``OutputFactor.toString``)
- ALS mean row        ``"MEAN,<U|I>,<f1>;...;<fk>"``
  (reference flink-als/.../ALSMeanVector.scala:35-40)
- Queryable-state key ``"<id>-U"`` / ``"<id>-I"``
  (reference als-ms/.../qs/ALSKafkaConsumer.java:79)
- SVM flat row        ``"<featureIndex>,<weight>"`` with 1-based index
  (reference flink-svm/.../SVMImpl.scala:33-35,45)
- SVM range row       ``"<bucket>,<i1>:<w1>;<i2>:<w2>;..."`` with
  ``bucket = featureIndex / range`` (1-based index, integer division;
  reference flink-svm/.../SVMImpl.scala:40-44,63-71)

``java_double_to_string`` reproduces ``Double.toString`` (JLS / JDK>=19 Ryu
semantics: shortest decimal that round-trips, decimal notation for
1e-3 <= |d| < 1e7, computerized scientific notation otherwise).  Python's
``repr`` provides the same shortest-round-trip digit selection; only the
surface formatting differs, which is what this module implements.

PARITY SCOPE: byte parity is guaranteed against JDK>=19 ``Double.toString``
(Ryu, JDK-4511638).  The reference stack is Flink-1.3-era Java 8, whose
legacy ``FloatingDecimal`` emits LONGER (non-shortest) digit strings for
some doubles — rows produced by an actual Java 8 reference job can differ
in those rare trailing digits.  Values still round-trip identically (every
consumer, reference included, re-parses with ``Double.parseDouble``), so
the numeric state contract holds; only byte-for-byte diffing of Java-8-
produced files against ours can show benign differences.
"""

from __future__ import annotations

import math
from decimal import Decimal
from typing import Iterable, List, Sequence, Tuple


def java_double_to_string(x: float) -> str:
    """Format ``x`` exactly as Java's ``Double.toString(double)`` would."""
    if math.isnan(x):
        return "NaN"
    if math.isinf(x):
        return "Infinity" if x > 0 else "-Infinity"
    if x == 0.0:
        return "-0.0" if math.copysign(1.0, x) < 0 else "0.0"

    sign = "-" if x < 0 else ""
    # Shortest round-trip digits via repr; Decimal parses them exactly.
    t = Decimal(repr(abs(x))).as_tuple()
    digits = "".join(map(str, t.digits))
    # pointpos = number of digits before the decimal point.
    pointpos = len(digits) + t.exponent
    digits = digits.rstrip("0") or "0"

    if -2 <= pointpos <= 7:  # 1e-3 <= |x| < 1e7  -> plain decimal notation
        if pointpos <= 0:
            return f"{sign}0.{'0' * (-pointpos)}{digits}"
        if pointpos >= len(digits):
            return f"{sign}{digits}{'0' * (pointpos - len(digits))}.0"
        return f"{sign}{digits[:pointpos]}.{digits[pointpos:]}"
    # scientific: d.dddE<exp>, exponent has no '+' / leading zeros
    mant_rest = digits[1:] or "0"
    return f"{sign}{digits[0]}.{mant_rest}E{pointpos - 1}"


def format_factors(factors: Sequence[float]) -> str:
    """``f1;f2;...;fk`` — Array[Double].mkString(";") with Java doubles."""
    return ";".join(java_double_to_string(float(f)) for f in factors)


def parse_factors(s: str) -> List[float]:
    return [float(tok) for tok in s.split(";")]


# ---------------------------------------------------------------- ALS rows

def als_factor_row(entity_id, kind: str, factors: Sequence[float]) -> str:
    """One ALS model row.  ``kind`` is "U" or "I"; id is an integral id or
    the literal "MEAN" (cold-start mean vector)."""
    if kind not in ("U", "I"):
        raise ValueError(f"kind must be 'U' or 'I', got {kind!r}")
    return f"{entity_id},{kind},{format_factors(factors)}"


def parse_als_row(row: str) -> Tuple[str, str, List[float]]:
    """Parse ``"<id>,<U|I>,<f;f;f>"`` -> (id, kind, factors).

    id stays a string: it is "MEAN" for mean rows and a decimal integer
    otherwise (the consumer keys state by the string form,
    reference ALSKafkaConsumer.java:73-82).
    """
    tokens = row.strip().split(",")
    if len(tokens) != 3:
        raise ValueError(f"bad ALS model row: {row!r}")
    entity_id, kind, facs = tokens
    if kind not in ("U", "I"):
        raise ValueError(f"bad ALS factor kind in row: {row!r}")
    return entity_id, kind, parse_factors(facs)


def als_state_key(entity_id, kind: str) -> str:
    """Queryable-state key ``"<id>-U"`` (ALSKafkaConsumer.java:79)."""
    return f"{entity_id}-{kind}"


MEAN_ID = "MEAN"


# ---------------------------------------------------------------- SVM rows

def svm_flat_row(index_1based: int, weight: float) -> str:
    return f"{index_1based},{java_double_to_string(float(weight))}"


def parse_svm_flat_row(row: str) -> Tuple[int, float]:
    idx, w = row.strip().split(",")
    return int(idx), float(w)


def svm_bucket_of(index_1based: int, range_size: int) -> int:
    """``bucket = featureIndex / range`` — integer division on the 1-based
    index (SVMImpl.scala:42; RangePartitionSVMPredict.java:63)."""
    return index_1based // range_size


def svm_range_row(bucket: int, pairs: Iterable[Tuple[int, float]]) -> str:
    """``"<bucket>,<i>:<w>;<i>:<w>"`` (SVMImpl.scala rangePartition:63-71)."""
    body = ";".join(
        f"{i}:{java_double_to_string(float(w))}" for i, w in pairs
    )
    return f"{bucket},{body}"


def parse_svm_range_row(row: str) -> Tuple[int, List[Tuple[int, float]]]:
    bucket_s, body = row.strip().split(",", 1)
    pairs: List[Tuple[int, float]] = []
    for item in body.split(";"):
        i, w = item.split(":")
        pairs.append((int(i), float(w)))
    return int(bucket_s), pairs


# ----------------------------------------------------- load-generator CSVs

def als_latency_csv_row(u_id: int, i_id: int, prediction: float, millis: float) -> str:
    """``uId,iId,prediction,millis`` (ALSPredictRandom.java:94,106-107)."""
    return f"{u_id},{i_id},{java_double_to_string(prediction)},{millis:.0f}"


def svm_latency_csv_row(query_id: int, nnz: int, prediction: float, millis: float) -> str:
    """``queryId,nnz,prediction,millis`` (SVMPredictRandom.java:91,97-98)."""
    return f"{query_id},{nnz},{java_double_to_string(prediction)},{millis:.0f}"
