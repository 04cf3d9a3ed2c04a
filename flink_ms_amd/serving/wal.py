"""Durable ingest journal: Kafka at-least-once parity for the serving state.

The reference's model stream survives consumer restarts because the rows
live in a Kafka topic (producer flush-on-checkpoint,
als-ms/.../qs/ALSKafkaProducer.java:36-37) and the consumer restores keyed
state from its checkpoint then re-reads the topic from the committed offset
(ALSKafkaConsumer.java:44-51).  The rebuild's equivalent under
``--stateBackend fs``: every ingested row (model rows, online-SGD
emissions) is appended to a write-ahead segment in ``<checkpointDataUri>/
wal/`` before the HTTP reply, and on restart the newest snapshot is
re-ingested followed by an in-order replay of all surviving segments.

Replay is idempotent by construction: rows are last-writer-wins upserts
(the Kafka contract), so re-applying rows the snapshot
already covers converges to the same state — which makes the
checkpoint/rotate crash windows safe (worst case: extra replay work).
Segments older than a completed snapshot are deleted at rotation.

Durability level matches Kafka's default: buffered writes flushed per
append (page cache), with ``fsync=True`` for per-append fsync.
"""

from __future__ import annotations

import os
import threading
from typing import Iterable, List, Optional


class IngestJournal:
    def __init__(self, directory: str, name: str, fsync: bool = False):
        self.dir = directory
        self.name = name
        self.fsync = fsync
        os.makedirs(directory, exist_ok=True)
        self._lock = threading.Lock()
        self._fh = None
        seqs = [int(f.split("-")[-1].split(".")[0])
                for f in self._segments()]
        self._seq = (max(seqs) + 1) if seqs else 0

    def _segments(self) -> List[str]:
        try:
            return sorted(f for f in os.listdir(self.dir)
                          if f.startswith(self.name + "-")
                          and f.endswith(".wal"))
        except FileNotFoundError:
            return []

    def _segment_path(self, seq: int) -> str:
        return os.path.join(self.dir, f"{self.name}-{seq:08d}.wal")

    def append(self, rows: Iterable[str]) -> int:
        """Append rows durably (called BEFORE the ingest reply)."""
        data = "".join(r.rstrip("\n") + "\n" for r in rows if r.strip())
        if not data:
            return 0
        with self._lock:
            if self._fh is None:
                self._fh = open(self._segment_path(self._seq), "a")
            self._fh.write(data)
            self._fh.flush()
            if self.fsync:
                os.fsync(self._fh.fileno())
        return data.count("\n")

    def rotate(self) -> None:
        """Called after a completed snapshot: everything in the journal is
        now covered by the snapshot, so seal the current segment, delete
        every closed segment, and start fresh.  A crash between the
        snapshot write and the deletes only causes idempotent re-replay."""
        with self._lock:
            if self._fh is not None:
                self._fh.close()
                self._fh = None
            current = self._segment_path(self._seq)
            for f in self._segments():
                path = os.path.join(self.dir, f)
                if path != current:
                    os.unlink(path)
            if os.path.exists(current):
                os.unlink(current)
            self._seq += 1

    def replay_rows(self) -> List[str]:
        """All journaled rows in append order (oldest segment first)."""
        rows: List[str] = []
        for f in self._segments():
            with open(os.path.join(self.dir, f)) as fh:
                rows.extend(ln for ln in fh.read().splitlines()
                            if ln.strip())
        return rows

    def close(self) -> None:
        with self._lock:
            if self._fh is not None:
                self._fh.close()
                self._fh = None


def journal_dir(checkpoint_data_uri: str) -> str:
    return os.path.join(checkpoint_data_uri, "wal")
