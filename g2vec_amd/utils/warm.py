"""Device-op warmup.

The first CUDA call of each distinct torch op loads its code object from
the fat binary (hipModule load + dispatcher init) — measured ~1.2 s
across the step-3 op set on a fresh process (profiles/step3_cold.json:
0.92 s in graph ops, 0.26 s in integrate ops, vs a 6 ms warmed step-3).
Running the same op set once on 64-gene toy tensors triggers the same
one-time loads in ~1 s of wall but off the measured path, so reported
phase timings attribute algorithm cost, not runtime initialization.
"""
from __future__ import annotations

import torch


def warm_device_ops(device: torch.device) -> None:
    """Exercise the step-3 + trainer op set on tiny tensors."""
    if device.type != "cuda":
        return
    from ..graph import build_group_graph
    from ..paths import integrate_pathsets
    from ..walks import generate_walks

    G = 64
    gen = torch.Generator().manual_seed(0)
    expr = torch.randn(16, G, generator=gen).to(device)
    labels = (torch.arange(16) % 2).to(device)
    e = torch.randint(0, G, (256, 2), generator=gen,
                      dtype=torch.int64).to(device)
    ws = []
    for grp in (0, 1):
        g = build_group_graph(expr, labels, grp, e, G)
        ws.append(generate_walks(g, 8, 2, 0, grp))
    integrate_pathsets(ws[0], ws[1], G)
    torch.cuda.synchronize()
