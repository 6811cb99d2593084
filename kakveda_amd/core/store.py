"""Append-only JSONL persistence, wire-compatible with the reference's data
files (data/failures.jsonl, data/patterns.jsonl, data/health.jsonl; see
/root/reference/services/gfkb/app.py:38-51 for the access pattern).

Unlike the reference (full-file reload on every query, no locking —
SURVEY.md section 5.2), this store keeps an in-memory view that is the
source of truth for reads, appends synchronously under a lock, and reloads
from disk only at construction. The GFKB GPU engine layers the HBM-resident
embedding mirror on top of this (gfkb/engine.py).
"""

from __future__ import annotations

import json
import os
import threading
from pathlib import Path
from typing import Any, Callable, Dict, Iterable, List, Optional


class JsonlLog:
    """A durable append-only log of JSON objects with an in-memory view."""

    def __init__(self, path: str | os.PathLike):
        self.path = Path(path)
        self._lock = threading.Lock()
        self._records: List[Dict[str, Any]] = []
        if self.path.exists():
            with self.path.open("r", encoding="utf-8") as fh:
                for line in fh:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        self._records.append(json.loads(line))
                    except json.JSONDecodeError:
                        continue  # tolerate a torn trailing line

    def __len__(self) -> int:
        return len(self._records)

    def all(self) -> List[Dict[str, Any]]:
        with self._lock:
            return list(self._records)

    def append(self, record: Dict[str, Any]) -> None:
        """Durably append one record (disk first, then the memory view)."""
        line = json.dumps(record, ensure_ascii=False, default=str)
        with self._lock:
            self.path.parent.mkdir(parents=True, exist_ok=True)
            with self.path.open("a", encoding="utf-8") as fh:
                fh.write(line + "\n")
            self._records.append(json.loads(line))

    def extend(self, records: Iterable[Dict[str, Any]]) -> None:
        for r in records:
            self.append(r)

    def tail(self, n: int, where: Optional[Callable[[Dict[str, Any]], bool]] = None) -> List[Dict[str, Any]]:
        with self._lock:
            if where is None:
                return list(self._records[-n:])
            out: List[Dict[str, Any]] = []
            for rec in reversed(self._records):
                if where(rec):
                    out.append(rec)
                    if len(out) >= n:
                        break
            return list(reversed(out))

    def rewrite(self, records: List[Dict[str, Any]], backup_suffix: Optional[str] = None) -> None:
        """Atomically replace the log (used by admin purge; the reference
        takes timestamped .bak- backups first, dashboard/app.py:318-327)."""
        with self._lock:
            if backup_suffix and self.path.exists():
                self.path.replace(self.path.with_name(self.path.name + backup_suffix))
            tmp = self.path.with_suffix(".tmp")
            with tmp.open("w", encoding="utf-8") as fh:
                for r in records:
                    fh.write(json.dumps(r, ensure_ascii=False, default=str) + "\n")
            tmp.replace(self.path)
            self._records = list(records)
