"""Global Failure Knowledge Base: HBM-resident embedding store + JSONL log."""

from kakveda_amd.gfkb.engine import EmbeddingStore, GfkbEngine  # noqa: F401
