"""The GFKB engine: durable JSONL log + device-resident embedding store.

Service-level parity with the reference GFKB
(/root/reference/services/gfkb/app.py:74-198): versioned append-only
failure upserts keyed by (failure_type, signature_text), F-%04d /
FP-%04d id numbering over total row count, top-5-then-type-filter match
ordering, latest-per-id pattern listing, name-identity pattern upserts.

MI355X-first redesign: instead of re-reading the
JSONL and refitting TF-IDF per request, the engine keeps one embedding row
per failure *identity* in a preallocated device tensor (bf16 in HBM3E on
GPU, fp32 on CPU) and matches with the fused cosine-topk kernel. Deliberate
divergence from the reference: the reference scores every *version* row, so
its top-5 can be five copies of one failure; this engine searches one row
per identity and reports the latest version — strictly more informative,
same wire shapes.
"""

from __future__ import annotations

import threading
from pathlib import Path
from typing import Any, Dict, List, Optional, Tuple

import torch

from kakveda_amd import ops
from kakveda_amd.core.schemas import (
    CanonicalFailureRecord,
    FailureMatch,
    PatternEntity,
    Severity,
    utcnow,
)
from kakveda_amd.core.store import JsonlLog
from kakveda_amd.encoder.model import TraceEncoder


class EmbeddingStore:
    """A growable [cap, D] row store on a device; rows are unit vectors.

    On GPU the tensor is bf16 in HBM3E; queries arrive fp32 and are cast.
    Insertion appends rows at the live prefix; the search kernel only scans
    the first ``count`` rows (valid_n), so inserts while serving are safe
    under the engine lock.
    """

    def __init__(self, dim: int, device: str = "cpu", capacity: int = 1024):
        self.dim = dim
        self.device = torch.device(device)
        self.dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self._data = torch.zeros(capacity, dim, dtype=self.dtype, device=self.device)
        self.count = 0

    @property
    def capacity(self) -> int:
        return self._data.shape[0]

    def _grow_to(self, needed: int) -> None:
        cap = self.capacity
        while cap < needed:
            cap *= 2
        if cap != self.capacity:
            fresh = torch.zeros(cap, self.dim, dtype=self.dtype, device=self.device)
            fresh[: self.count] = self._data[: self.count]
            self._data = fresh

    def adopt(self, data: torch.Tensor) -> None:
        """Take ownership of a pre-built [n, D] row tensor (zero copy) —
        used by benches/restore paths where the rows already live on the
        device in the right dtype."""
        assert data.dtype == self.dtype and data.shape[1] == self.dim
        self._data = data
        self.count = data.shape[0]

    def append(self, rows: torch.Tensor) -> int:
        """Append [n, D] rows; returns the first new row index."""
        n = rows.shape[0]
        self._grow_to(self.count + n)
        first = self.count
        self._data[first : first + n] = rows.to(self.dtype)
        self.count += n
        return first

    def search(self, queries: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
        return ops.cosine_topk(queries.to(self.dtype), self._data, k, valid_n=self.count)

    @property
    def data(self) -> torch.Tensor:
        return self._data


class GfkbEngine:
    def __init__(
        self,
        data_dir: str,
        device: str = "cpu",
        dim: int = 768,
        hash_dim: int = 1 << 16,
        encoder_seed: int = 1234,
        top_k: int = 5,
    ):
        self.data_dir = Path(data_dir)
        self.data_dir.mkdir(parents=True, exist_ok=True)
        self.failures = JsonlLog(self.data_dir / "failures.jsonl")
        self.patterns = JsonlLog(self.data_dir / "patterns.jsonl")
        self.top_k = top_k
        self._lock = threading.RLock()

        self.encoder = TraceEncoder(dim=dim, hash_dim=hash_dim, seed=encoder_seed, device=device)
        self.store = EmbeddingStore(dim, device=device, capacity=1024)

        # identity (failure_type, signature_text) -> latest record dict
        self._latest: Dict[Tuple[str, str], Dict[str, Any]] = {}
        # store row -> identity
        self._row_identity: List[Tuple[str, str]] = []
        self._identity_row: Dict[Tuple[str, str], int] = {}
        self._rebuild_from_log()

    # -- bootstrap ---------------------------------------------------------

    def _rebuild_from_log(self) -> None:
        """Rebuild the HBM mirror from the durable JSONL."""
        for rec in self.failures.all():
            key = (rec.get("failure_type", ""), rec.get("signature_text", ""))
            self._latest[key] = rec
        identities = list(self._latest.keys())
        if identities:
            emb = self.encoder.encode_texts([sig for (_ft, sig) in identities])
            self.store.append(emb)
            self._row_identity = identities
            self._identity_row = {key: i for i, key in enumerate(identities)}

    def attach_store(self, new_store) -> None:
        """Swap the embedding store (e.g. for a distributed coordinator
        store) and re-encode every known identity into it, preserving row
        order. Must happen before serving begins."""
        with self._lock:
            self.store = new_store
            if self._row_identity:
                emb = self.encoder.encode_texts(
                    [sig for (_ft, sig) in self._row_identity]
                )
                self.store.append(emb)

    # -- failures ----------------------------------------------------------

    def list_failures(self) -> List[Dict[str, Any]]:
        return self.failures.all()

    def upsert_failure(
        self,
        failure_type: str,
        signature_text: str,
        context_signature: Dict[str, Any],
        impact_severity: str = "medium",
        root_cause: Optional[str] = None,
        resolution: Optional[str] = None,
        app_id: str = "",
    ) -> Tuple[Dict[str, Any], bool]:
        """Versioned upsert; returns (record, created)."""
        with self._lock:
            key = (failure_type, signature_text)
            existing = self._latest.get(key)
            now = utcnow()
            if existing is None:
                rec = CanonicalFailureRecord(
                    failure_id=f"F-{len(self.failures) + 1:04d}",
                    version=1,
                    created_at=now,
                    updated_at=now,
                    failure_type=failure_type,
                    root_cause=root_cause,
                    context_signature=context_signature,
                    impact_severity=Severity(impact_severity),
                    resolution=resolution,
                    occurrences=1,
                    affected_apps=[app_id] if app_id else [],
                    signature_text=signature_text,
                ).model_dump(mode="json")
                self.failures.append(rec)
                self._latest[key] = rec
                emb = self.encoder.encode_texts([signature_text])
                row = self.store.append(emb)
                self._row_identity.append(key)
                self._identity_row[key] = row
                return rec, True

            rec = dict(existing)
            rec["version"] = int(rec["version"]) + 1
            rec["updated_at"] = now.isoformat().replace("+00:00", "Z")
            rec["occurrences"] = int(rec.get("occurrences", 0)) + 1
            apps = list(rec.get("affected_apps", []))
            if app_id and app_id not in apps:
                apps.append(app_id)
            rec["affected_apps"] = apps
            rec["root_cause"] = root_cause or rec.get("root_cause")
            rec["resolution"] = resolution or rec.get("resolution")
            rec["context_signature"] = context_signature or rec.get("context_signature")
            self.failures.append(rec)
            self._latest[key] = rec
            return rec, False

    def match(
        self, signature_text: str, failure_type: Optional[str] = None, top_k: Optional[int] = None
    ) -> List[FailureMatch]:
        """Top-k failures by cosine similarity, then optional type filter
        (reference order: cut to top-k first, filter second)."""
        k = top_k or self.top_k
        with self._lock:
            if self.store.count == 0:
                return []
            q = self.encoder.encode_texts([signature_text])
            scores, idx = self.store.search(q, min(k, self.store.count))
        out: List[FailureMatch] = []
        for s, i in zip(scores[0].tolist(), idx[0].tolist()):
            if i < 0 or i >= len(self._row_identity):
                # rows the engine has no identity for (e.g. a pre-loaded
                # corpus adopted under the engine) can win the similarity
                # search but cannot be reported as failures
                continue
            rec = self._latest[self._row_identity[i]]
            if failure_type and rec["failure_type"] != failure_type:
                continue
            out.append(
                FailureMatch(
                    failure_id=rec["failure_id"],
                    version=rec["version"],
                    score=float(s),
                    failure_type=rec["failure_type"],
                    suggested_mitigation=rec.get("resolution"),
                )
            )
        return out

    # -- patterns ----------------------------------------------------------

    def list_patterns(self) -> List[Dict[str, Any]]:
        latest: Dict[str, Dict[str, Any]] = {}
        for rec in self.patterns.all():
            latest[rec.get("pattern_id") or rec.get("name", "")] = rec
        return list(latest.values())

    def upsert_pattern(
        self,
        name: str,
        failure_ids: List[str],
        affected_apps: List[str],
        description: Optional[str] = None,
    ) -> Tuple[Dict[str, Any], bool]:
        with self._lock:
            existing = None
            for rec in reversed(self.patterns.all()):
                if rec.get("name") == name:
                    existing = rec
                    break
            if existing is None:
                rec = PatternEntity(
                    pattern_id=f"FP-{len(self.patterns) + 1:04d}",
                    name=name,
                    created_at=utcnow(),
                    failure_ids=sorted(set(failure_ids)),
                    affected_apps=sorted(set(affected_apps)),
                    description=description,
                ).model_dump(mode="json")
                self.patterns.append(rec)
                return rec, True
            rec = dict(existing)
            rec["failure_ids"] = sorted(set(list(rec.get("failure_ids", [])) + failure_ids))
            rec["affected_apps"] = sorted(set(list(rec.get("affected_apps", [])) + affected_apps))
            rec["description"] = description or rec.get("description")
            self.patterns.append(rec)
            return rec, False
