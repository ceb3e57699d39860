"""Device normalisation.

``torch.device("cuda") != torch.device("cuda:0")`` even when they refer
to the same GPU, so any "is this tensor already on my device" cache
check silently fails for an index-less device string — observed as the
serving scorer re-allocating its coefficient tensor on EVERY predict,
which aborts hipGraph capture ("operation not permitted when stream is
capturing") and would desynchronise captured graphs from hot-redeploy
weight swaps.  Canonicalise once at construction instead.
"""
from __future__ import annotations

import torch


def canonical_device(device) -> torch.device:
    d = torch.device(device)
    if d.type == "cuda" and d.index is None:
        idx = torch.cuda.current_device() if torch.cuda.is_available() else 0
        d = torch.device("cuda", idx)
    return d
