"""kakveda-amd: an MI355X-native LLM failure-intelligence engine.

A from-scratch re-design of the capabilities of prateekdevisingh/kakveda
(reference layout surveyed in SURVEY.md) built GPU-first for AMD MI355X
(gfx950 / CDNA4):

- The Global Failure Knowledge Base (GFKB) is an HBM3E-resident
  fingerprint-embedding store (``kakveda_amd.gfkb``) searched by a
  hand-written MFMA-tiled cosine top-k HIP kernel (``kakveda_amd.ops``).
- Pre-flight warning checks, pattern clustering (streaming k-means) and the
  trace encoder run on GPU; multi-GPU sharding uses RCCL over xGMI via
  ``torch.distributed`` (``kakveda_amd.parallel``).
- The HTTP/event microservice surface (``kakveda_amd.services``) stays
  wire-compatible with the reference (same endpoints, topics, JSONL record
  shapes), so existing clients keep working.

Reference parity map: see SURVEY.md section 2 and the docstrings of each
submodule, which cite the reference files they are behaviourally equivalent
to (e.g. services/shared/models.py, services/gfkb/app.py).
"""

__version__ = "0.1.0"
