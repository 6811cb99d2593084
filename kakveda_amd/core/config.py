"""Hot-reloading YAML configuration store.

Parity with the reference's ConfigStore
(/root/reference/services/shared/config.py:18-58): one YAML file shared by
all services, re-read when its mtime changes, with a minimum poll interval
governed by ``hot_reload.poll_seconds`` in the file itself.
"""

from __future__ import annotations

import copy
import os
import time
from pathlib import Path
from typing import Any, Dict, Optional

import yaml

DEFAULT_CONFIG: Dict[str, Any] = {
    "failure_matching": {"similarity_threshold": 0.8},
    "warning_policy": {"default_action": "warn"},
    "health_score": {
        "severity_weights": {"low": 1, "medium": 3, "high": 7},
        "base_score": 100,
        "window_size": 10,
    },
    "sampling": {"enabled": False},
    "hot_reload": {"enabled": True, "poll_seconds": 2},
    "gfkb": {
        "embedding_dim": 768,
        "top_k": 5,
        "hash_dim": 1 << 16,
        "encoder_seed": 1234,
    },
}


class ConfigStore:
    """Reads a YAML config file and hot-reloads on mtime change."""

    def __init__(self, path: Optional[str] = None):
        self.path = Path(path or os.environ.get("CONFIG_PATH", "/app/config/config.yaml"))
        self._data: Dict[str, Any] = copy.deepcopy(DEFAULT_CONFIG)
        self._mtime: float = -1.0
        self._last_check: float = 0.0
        self._load(force=True)

    def _load(self, force: bool = False) -> None:
        now = time.monotonic()
        hot = self._data.get("hot_reload", {}) or {}
        raw_poll = hot.get("poll_seconds", 2)
        poll = 2.0 if raw_poll is None else float(raw_poll)  # 0 = poll always
        if not force:
            if not bool(hot.get("enabled", True)):
                return
            if now - self._last_check < poll:
                return
        self._last_check = now
        try:
            mtime = self.path.stat().st_mtime
        except OSError:
            return
        if not force and mtime == self._mtime:
            return
        try:
            loaded = yaml.safe_load(self.path.read_text()) or {}
        except Exception:
            return
        if isinstance(loaded, dict):
            merged = copy.deepcopy(DEFAULT_CONFIG)  # deep: _deep_update must
            # never mutate the shared defaults
            _deep_update(merged, loaded)
            self._data = merged
            self._mtime = mtime

    def get(self, dotted: str, default: Any = None) -> Any:
        """Fetch ``"a.b.c"`` style keys, hot-reloading first."""
        self._load()
        node: Any = self._data
        for part in dotted.split("."):
            if not isinstance(node, dict) or part not in node:
                return default
            node = node[part]
        return node

    def snapshot(self) -> Dict[str, Any]:
        self._load()
        return dict(self._data)


def _deep_update(dst: Dict[str, Any], src: Dict[str, Any]) -> None:
    for k, v in src.items():
        if isinstance(v, dict) and isinstance(dst.get(k), dict):
            _deep_update(dst[k], v)
        else:
            dst[k] = v
