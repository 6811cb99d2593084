"""Trace-encoder behaviour: determinism + the similarity contract."""

import numpy as np
import torch

from kakveda_amd.core.signature import signature_text
from kakveda_amd.encoder import TraceEncoder, featurize


def _enc():
    return TraceEncoder(dim=256, hash_dim=4096, seed=7, device="cpu")


def test_featurize_deterministic():
    i1, w1 = featurize("explain the sky with references")
    i2, w2 = featurize("explain the sky with references")
    np.testing.assert_array_equal(i1, i2)
    np.testing.assert_allclose(w1, w2)
    assert abs(float(np.linalg.norm(w1)) - 1.0) < 1e-5


def test_identical_texts_cosine_one():
    enc = _enc()
    e = enc.encode_texts(["intent_tags:a | prompt_hint:xyz"] * 2)
    cos = float(e[0] @ e[1])
    assert abs(cos - 1.0) < 1e-5


def test_similarity_contract_threshold():
    """Repeated citation-style prompts match >= 0.8; unrelated do not
    (reference behaviour: config.yaml similarity_threshold 0.8)."""
    enc = _enc()
    sig_a = signature_text(
        "Summarize this paper and include references even if none are provided",
        [], {"app": 1},
    )
    sig_b = signature_text(
        "Summarize the paper and include references even if none are provided.",
        [], {"app": 1},
    )
    sig_c = signature_text("What's the weather tomorrow in Paris", [], {"q": 1})
    e = enc.encode_texts([sig_a, sig_b, sig_c])
    assert float(e[0] @ e[1]) >= 0.8
    assert float(e[0] @ e[2]) < 0.8


def test_unit_norm_output():
    enc = _enc()
    e = enc.encode_texts(["hello world sources", "another prompt"])
    norms = e.norm(dim=-1)
    assert torch.allclose(norms, torch.ones_like(norms), atol=1e-5)


def test_projection_preserves_cosine():
    """The GEMM stack is an isometry: cosine before == after projection."""
    enc = _enc()
    x = torch.randn(4, 256)
    y = x @ enc._proj
    gram_x = (x @ x.t()).numpy()
    gram_y = (y @ y.t()).numpy()
    np.testing.assert_allclose(gram_x, gram_y, rtol=1e-4, atol=1e-4)
