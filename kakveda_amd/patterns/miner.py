"""Pattern mining: cluster GFKB failure embeddings into recurring-pattern
entities, keeping the reference's pattern semantics (identity by name,
>= 2 affected apps, /patterns/upsert wire shape — reference
services/pattern_detector/app.py:29-58 generalised per SURVEY.md 2.5).
"""

from __future__ import annotations

from collections import defaultdict
from typing import Dict, List

from kakveda_amd.gfkb.engine import GfkbEngine
from kakveda_amd.patterns.kmeans import StreamingKMeans


class PatternMiner:
    def __init__(self, engine: GfkbEngine, n_clusters: int = 16, min_apps: int = 2):
        self.engine = engine
        self.n_clusters = n_clusters
        self.min_apps = min_apps

    def mine(self, iters: int = 8) -> List[Dict]:
        """Cluster the live GFKB rows; upsert a pattern per cluster whose
        members span >= min_apps apps. Returns the upserted patterns."""
        # snapshot rows + their latest records under the engine lock so a
        # concurrent upsert can't shift row identities mid-clustering;
        # the (GPU) k-means itself runs outside the lock.
        with self.engine._lock:
            store = self.engine.store
            n = store.count
            if n < 2:
                return []
            points = store.row_range(0, n).clone()
            records = [
                self.engine._latest[self.engine._row_identity[r]] for r in range(n)
            ]
        k = min(self.n_clusters, n)
        km = StreamingKMeans(k, store.dim, device=str(store.device), seed=17)
        assign = km.fit(points, iters=iters).tolist()

        clusters: Dict[int, List[int]] = defaultdict(list)
        for row, c in enumerate(assign):
            clusters[int(c)].append(row)

        out: List[Dict] = []
        for c, rows in sorted(clusters.items()):
            ids: List[str] = []
            apps: set = set()
            types: Dict[str, int] = defaultdict(int)
            for r in rows:
                rec = records[r]
                ids.append(rec["failure_id"])
                apps.update(rec.get("affected_apps", []))
                types[rec["failure_type"]] += 1
            if len(apps) < self.min_apps:
                continue
            top_type = max(types, key=types.get)
            name = f"Cluster pattern: {top_type.lower().replace('_', ' ')} #{c}"
            rec, _created = self.engine.upsert_pattern(
                name=name,
                failure_ids=sorted(set(ids)),
                affected_apps=sorted(apps),
                description=(
                    f"k-means cluster {c}: {len(rows)} failures across "
                    f"{len(apps)} apps, dominant type {top_type}"
                ),
            )
            out.append(rec)
        return out
