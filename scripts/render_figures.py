"""Regenerate docs/figures SVGs (parity with reference scripts/render_figures.py).

Pure-stdlib SVG writer: the service-constellation diagram and the GPU
data-plane diagram referenced from the docs.
"""

from __future__ import annotations

import os

OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "docs", "figures")


def _box(x, y, w, h, label, fill="#161b22"):
    return (
        f'<rect x="{x}" y="{y}" width="{w}" height="{h}" rx="8" fill="{fill}" '
        f'stroke="#30363d"/>\n<text x="{x + w / 2}" y="{y + h / 2 + 4}" '
        f'text-anchor="middle" fill="#e6edf3" font-size="12" '
        f'font-family="monospace">{label}</text>'
    )


def _arrow(x1, y1, x2, y2):
    return (
        f'<line x1="{x1}" y1="{y1}" x2="{x2}" y2="{y2}" stroke="#58a6ff" '
        'stroke-width="1.5" marker-end="url(#a)"/>'
    )


HEAD = (
    '<svg xmlns="http://www.w3.org/2000/svg" width="{w}" height="{h}" '
    'viewBox="0 0 {w} {h}"><defs><marker id="a" viewBox="0 0 10 10" '
    'refX="9" refY="5" markerWidth="7" markerHeight="7" orient="auto-start-reverse">'
    '<path d="M 0 0 L 10 5 L 0 10 z" fill="#58a6ff"/></marker></defs>'
    '<rect width="{w}" height="{h}" fill="#0d1117"/>'
)


def services_svg() -> str:
    parts = [HEAD.format(w=760, h=420)]
    boxes = {
        "client": (20, 30, 110, 36, "client / agent"),
        "warn": (320, 30, 150, 36, "warning_policy :8105"),
        "gfkb": (560, 30, 160, 36, "gfkb :8101 (GPU)"),
        "ingest": (20, 130, 140, 36, "ingestion :8102"),
        "bus": (240, 130, 140, 36, "event_bus :8100"),
        "fc": (460, 130, 200, 36, "failure_classifier :8103"),
        "pd": (140, 230, 190, 36, "pattern_detector :8104"),
        "hs": (400, 230, 180, 36, "health_scoring :8106"),
        "dash": (240, 330, 180, 36, "dashboard :8110"),
        "echo": (500, 330, 160, 36, "agent_echo :8120"),
    }
    for x, y, w, h, label in boxes.values():
        parts.append(_box(x, y, w, h, label))
    parts += [
        _arrow(130, 48, 320, 48),   # client -> warn
        _arrow(470, 48, 560, 48),   # warn -> gfkb
        _arrow(90, 66, 90, 130),    # client -> ingestion
        _arrow(160, 148, 240, 148),
        _arrow(380, 148, 460, 148),  # bus -> classifier (trace.ingested)
        _arrow(560, 130, 620, 66),   # classifier -> gfkb upsert
        _arrow(310, 166, 240, 230),  # failure.detected -> pattern detector
        _arrow(340, 166, 470, 230),  # -> health scoring
        _arrow(310, 166, 330, 330),  # trace.ingested -> dashboard
        _arrow(330, 266, 560, 62),   # pattern upsert -> gfkb
    ]
    parts.append(
        '<text x="20" y="405" fill="#8b949e" font-size="11" font-family="monospace">'
        "topics: trace.ingested, failure.detected, child_safety_alert</text>"
    )
    parts.append("</svg>")
    return "\n".join(parts)


def dataplane_svg() -> str:
    parts = [HEAD.format(w=760, h=300)]
    parts.append(_box(20, 30, 200, 40, "signature text"))
    parts.append(_box(20, 120, 200, 50, "hashed n-grams +\nembedding_bag (HIP)"))
    parts.append(_box(20, 220, 200, 40, "768-d unit fingerprint"))
    parts.append(_box(300, 30, 200, 60, "GFKB shard / GPU\nbf16 NxD in HBM3E"))
    parts.append(_box(300, 130, 200, 60, "fused MFMA cosine\ntop-k kernel"))
    parts.append(_box(560, 80, 180, 50, "RCCL all-gather\n(score, id) / xGMI"))
    parts.append(_box(560, 180, 180, 50, "top-k merge +\nthreshold policy"))
    for a in [
        _arrow(120, 70, 120, 120),
        _arrow(120, 170, 120, 220),
        _arrow(220, 240, 360, 190),
        _arrow(400, 90, 400, 130),
        _arrow(500, 155, 560, 110),
        _arrow(650, 130, 650, 180),
    ]:
        parts.append(a)
    parts.append("</svg>")
    return "\n".join(parts)


def main() -> None:
    os.makedirs(OUT, exist_ok=True)
    for name, svg in (("services.svg", services_svg()), ("dataplane.svg", dataplane_svg())):
        path = os.path.join(OUT, name)
        with open(path, "w") as fh:
            fh.write(svg)
        print("wrote", path)


if __name__ == "__main__":
    main()
