"""The trace-encoder model: hashed features -> 768-d unit embedding.

MI355X-native analogue of the reference's similarity space (SURVEY.md
section 2.5 row 'failure_classifier.on_trace'): a random-init embedding
table (the hashed-feature projection) followed by a small GEMM stack whose
weight is orthogonal, so cosine structure of the sparse feature space is
preserved exactly — required to keep the reference's >=0.8 match-threshold
semantics (config/config.yaml:2) meaningful after encoding.

GPU path: fused embedding-bag HIP kernel (ops.embedding_bag) + hipBLASLt
GEMM via torch.matmul for the projection + fused L2-normalise. CPU path:
the same math in plain torch (deterministic, used by BASELINE config 1).
"""

from __future__ import annotations

from typing import Sequence

import torch

from kakveda_amd.encoder.featurizer import featurize_batch


class TraceEncoder:
    def __init__(
        self,
        dim: int = 768,
        hash_dim: int = 1 << 16,
        seed: int = 1234,
        device: str = "cpu",
        depth: int = 2,
        max_features: int = 64,
    ):
        self.dim = dim
        self.hash_dim = hash_dim
        self.seed = seed
        self.max_features = max_features
        self.device = torch.device(device)

        gen = torch.Generator(device="cpu").manual_seed(seed)
        # Hashed-feature embedding table, scaled so bag outputs are O(1).
        table = torch.randn(hash_dim, dim, generator=gen, dtype=torch.float32) / (dim**0.5)
        # GEMM stack: product of `depth` orthogonal matrices (exact isometry).
        w = torch.eye(dim, dtype=torch.float32)
        for _ in range(depth):
            q, r = torch.linalg.qr(torch.randn(dim, dim, generator=gen, dtype=torch.float32))
            q = q * torch.sign(torch.diagonal(r)).unsqueeze(0)  # fix QR sign ambiguity
            w = w @ q
        self._table = table.to(self.device)
        self._proj = w.to(self.device)
        if self.device.type == "cuda":
            # bf16 resident copies for the GPU hot path
            self._table_bf16 = self._table.to(torch.bfloat16)
            self._proj_bf16 = self._proj.to(torch.bfloat16)

    @torch.no_grad()
    def encode_texts(self, texts: Sequence[str]) -> torch.Tensor:
        """Encode signature texts -> [B, dim] unit-norm float32 embeddings."""
        idx, w = featurize_batch(
            texts, hash_dim=self.hash_dim, seed=0, max_features=self.max_features
        )
        return self.encode_features(
            torch.from_numpy(idx).to(self.device),
            torch.from_numpy(w).to(self.device),
        )

    @torch.no_grad()
    def encode_features(self, idx: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
        """Encode pre-hashed padded features [B, L] -> [B, dim] unit fp32."""
        if self.device.type == "cuda":
            from kakveda_amd import ops

            bag = ops.embedding_bag(self._table_bf16, idx, w)  # [B, dim] f32
            out = (bag.to(torch.bfloat16) @ self._proj_bf16).float()
        else:
            flat = self._table[idx.reshape(-1).long()].reshape(*idx.shape, self.dim)
            bag = (flat * w.unsqueeze(-1)).sum(dim=1)
            out = bag @ self._proj
        norm = out.norm(dim=-1, keepdim=True).clamp_min(1e-12)
        return out / norm

    @torch.no_grad()
    def encode_text(self, text: str) -> torch.Tensor:
        return self.encode_texts([text])[0]
