"""Key-partitioned serving across processes.

The reference's model is sharded: Flink hash-partitions keyed state across
TaskManagers (``keyBy(0).asQueryableState``, ALSKafkaConsumer.java:91-92)
and the client resolves a key's location before the KvState fetch
(QueryClientHelper.java:121: ``getKvState(jobId, name, key.hashCode(), ..)``).

Here the equivalent is N independent serving processes (each a full
`serving.app` instance with its own store) and CLIENT-SIDE routing by a
stable key hash — the JobManager location lookup collapses into
``shard_of(key, N)``.  The producer routes model rows the same way, so each
shard holds exactly its key range.  One GPU serves all local shard
processes (stores share the device), or shards can spread over the node's
8 GPUs via ``--device`` per shard.
"""

from __future__ import annotations

import zlib
from typing import Dict, List, Optional, Tuple

from .client import QueryClientHelper


def shard_of(key: str, num_shards: int) -> int:
    """Stable key -> shard map (crc32; the contract every publisher and
    client shares, like Flink's key-group assignment)."""
    if num_shards <= 1:
        return 0
    return zlib.crc32(key.encode()) % num_shards


def als_row_key(row: str) -> str:
    """State key of an ALS model row (ALSKafkaConsumer map semantics)."""
    toks = row.split(",", 2)
    return f"{toks[0]}-{toks[1]}"


def svm_row_key(row: str) -> str:
    return row.split(",", 1)[0]


class ShardedQueryClient:
    """Routes queryable-state calls across shard endpoints by key hash."""

    def __init__(self, endpoints: List[Tuple[str, int]],
                 query_timeout_s: float = 5.0):
        self.clients = [QueryClientHelper(h, p, query_timeout_s)
                        for h, p in endpoints]
        self.n = len(self.clients)

    def _c(self, key: str) -> QueryClientHelper:
        return self.clients[shard_of(key, self.n)]

    def query_state(self, name: str, key: str) -> Optional[Tuple[str, str]]:
        return self._c(key).query_state(name, key)

    def als_predict(self, user: str, item: str) -> dict:
        """Client-side dot of two routed lookups (the reference client's
        structure: 2 KvState fetches + local math, ALSPredict.java:69-83)."""
        u = self.query_state("ALS_MODEL", f"{user}-U")
        v = self.query_state("ALS_MODEL", f"{item}-I")
        if u is None or v is None:
            return {"found": False,
                    "message": "User or Item Factors do not exist in the "
                               f"model for the query: {user},{item}"}
        uf = [float(x) for x in u[1].split(";")]
        vf = [float(x) for x in v[1].split(";")]
        pred = sum(a * b for a, b in zip(uf, vf))
        return {"found": True, "prediction": pred}

    def svm_predict(self, vector: str, output_decision_function: bool = False,
                    threshold_value: float = 0.0) -> dict:
        raw = 0.0
        messages = []
        for tok in vector.strip().split():
            fid, val = tok.split(":")
            hit = self.query_state("SVM_MODEL", fid)
            if hit is None:
                messages.append(
                    f"Could not find the value for feature ID: {fid} ")
            else:
                raw += float(hit[1]) * float(val)
        pred = raw if output_decision_function else (
            1.0 if raw > threshold_value else -1.0)
        return {"prediction": pred, "raw": raw, "messages": messages}

    def ingest_rows(self, model: str, rows: List[str]) -> int:
        """Producer path: route each model row to its key's shard."""
        key_of = als_row_key if model == "als" else svm_row_key
        by_shard: Dict[int, List[str]] = {}
        for row in rows:
            row = row.strip()
            if row:
                by_shard.setdefault(shard_of(key_of(row), self.n),
                                    []).append(row)
        return sum(self.clients[s].ingest_rows(model, shard_rows)
                   for s, shard_rows in by_shard.items())

    def close(self):
        for c in self.clients:
            c.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class ShardedKvClient:
    """Key-routed point lookups over N native KvState servers (raw
    keep-alive sockets; the Netty-KvStateServer-per-TaskManager topology
    with `shard_of` as the location lookup).  ALS predict does its two
    routed state fetches and the dot client-side, exactly like the
    reference client (ALSPredict.java:69-83) — the shards may hold
    disjoint key ranges, so the server-side /als/predict shortcut only
    applies at one shard."""

    def __init__(self, ports: List[int], host: str = "127.0.0.1",
                 timeout_s: float = 5.0):
        import socket
        self._socks = []
        self._bufs = []
        for p in ports:
            s = socket.create_connection((host, p), timeout=timeout_s)
            s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            self._socks.append(s)
            self._bufs.append(b"")
        self.n = len(ports)

    def _get(self, shard: int, path: str):
        import json
        s = self._socks[shard]
        s.sendall(f"GET {path} HTTP/1.1\r\nHost: x\r\n\r\n".encode())
        buf = self._bufs[shard]
        while b"\r\n\r\n" not in buf:
            buf += s.recv(65536)
        head, rest = buf.split(b"\r\n\r\n", 1)
        cl = int([ln for ln in head.split(b"\r\n")
                  if ln.lower().startswith(b"content-length")][0]
                 .split(b":")[1])
        while len(rest) < cl:
            rest += s.recv(65536)
        self._bufs[shard] = rest[cl:]
        status = int(head.split(b" ", 2)[1])
        return status, json.loads(rest[:cl])

    def query_state(self, name: str, key: str) -> Optional[Tuple[str, str]]:
        st, body = self._get(shard_of(key, self.n),
                             f"/state/{name}/{key}")
        if st != 200:
            return None
        return tuple(body["value"])

    def als_predict(self, user: str, item: str) -> dict:
        u = self.query_state("ALS_MODEL", f"{user}-U")
        v = self.query_state("ALS_MODEL", f"{item}-I")
        if u is None or v is None:
            return {"found": False,
                    "message": "User or Item Factors do not exist in the "
                               f"model for the query: {user},{item}"}
        uv = [float(x) for x in u[1].split(";")]
        vv = [float(x) for x in v[1].split(";")]
        return {"found": True,
                "prediction": sum(a * b for a, b in zip(uv, vv))}

    def close(self):
        for s in self._socks:
            try:
                s.close()
            except OSError:
                pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
