"""Trace encoder: signature text -> 768-d fingerprint embedding.

Replaces the reference's per-query TF-IDF refit
(/root/reference/services/shared/similarity.py:10-20) with a fixed,
deterministic hashed-feature encoder whose output lives in a dense space
that the GPU cosine-kNN kernel can search without per-query fitting.
"""

from kakveda_amd.encoder.featurizer import featurize, featurize_batch  # noqa: F401
from kakveda_amd.encoder.model import TraceEncoder  # noqa: F401
