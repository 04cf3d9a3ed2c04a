"""Query client: the QueryClientHelper rebuild over HTTP.

Reference: flink-queryable-client/.../QueryClientHelper.java (duplicated in
als-ms/.../utils/): synchronous point lookups of keyed state with a timeout,
``Optional.empty`` for unknown keys.  Here the transport is HTTP against the
serving app instead of Akka+Netty KvState.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import httpx


class QueryClientHelper:
    """Synchronous state queries (QueryClientHelper.java:104-139 parity)."""

    def __init__(self, host: str = "localhost", port: int = 6123,
                 query_timeout_s: float = 5.0):
        self.base = f"http://{host}:{port}"
        self._client = httpx.Client(timeout=query_timeout_s)

    def query_state(self, name: str, key: str
                    ) -> Optional[Tuple[str, str]]:
        """Returns the Tuple2 (key, payload) or None (Optional.empty)."""
        r = self._client.get(f"{self.base}/state/{name}/{key}")
        if r.status_code == 404:
            return None
        r.raise_for_status()
        v = r.json()["value"]
        return (v[0], v[1])

    def als_predict(self, user: str, item: str) -> dict:
        r = self._client.get(f"{self.base}/als/predict",
                             params={"user": user, "item": item})
        r.raise_for_status()
        return r.json()

    def svm_predict(self, vector: str, output_decision_function: bool = False,
                    threshold_value: float = 0.0,
                    range_size: Optional[int] = None) -> dict:
        r = self._client.post(f"{self.base}/svm/predict", json={
            "vector": vector,
            "output_decision_function": output_decision_function,
            "threshold_value": threshold_value,
            "range": range_size,
        })
        r.raise_for_status()
        return r.json()

    def ingest_rows(self, model: str, rows: List[str],
                    retries: int = 3) -> int:
        """At-least-once publish (producer parity: setFlushOnCheckpoint,
        ALSKafkaProducer.java:35-37): retry on transport failure; ingest is
        idempotent (last writer wins per key)."""
        last: Optional[Exception] = None
        for _ in range(retries + 1):
            try:
                r = self._client.post(f"{self.base}/model/{model}/rows",
                                      json={"rows": rows})
                r.raise_for_status()
                return r.json()["ingested"]
            except httpx.TransportError as e:
                last = e
        raise last

    def sgd_update(self, ratings: List[str], **kw) -> dict:
        payload = {"ratings": ratings}
        payload.update(kw)
        r = self._client.post(f"{self.base}/sgd/update", json=payload)
        r.raise_for_status()
        return r.json()

    def sgd_update_batch(self, ratings, field_delimiter: str = "\t",
                         learning_rate: float = 0.1,
                         user_regularization: float = 0.0,
                         item_regularization: float = 0.0) -> dict:
        """Batched online SGD through the K4 device kernel
        (/sgd/update_batch); unresolvable ids fall back to the scalar
        MEAN-cold-start path server-side."""
        r = self._client.post(f"{self.base}/sgd/update_batch", json={
            "ratings": ratings,
            "field_delimiter": field_delimiter,
            "learning_rate": learning_rate,
            "user_regularization": user_regularization,
            "item_regularization": item_regularization,
        })
        r.raise_for_status()
        return r.json()

    def close(self):
        self._client.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
