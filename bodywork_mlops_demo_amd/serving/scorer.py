"""Batched, hipGraph-captured inference for the scoring service.

The reference scores one row per HTTP request through sklearn
``model.predict`` (``stage_2_serve_model.py:73-80``).  Here the scorer
keeps the model resident in HBM, scores arbitrary batches through the HIP
kernels, and replays hipGraph-captured launches for the common batch
sizes so the per-request launch overhead disappears from the hot loop
(SURVEY.md §7 step 4).

Graph strategy: power-of-two batch buckets, one captured graph per
bucket, static input/output buffers; a request is padded up to its
bucket, replayed, and sliced.  Capture is lazy (first use of a bucket)
and falls back to direct launches on CPU or when capture is unavailable.
"""
from __future__ import annotations

import numpy as np
import torch

from bodywork_mlops_demo_amd.utils.device import canonical_device

from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)


class BatchedScorer:
    BUCKETS = [1, 16, 256, 4096, 65536, 1 << 20, 1 << 22, 1 << 24]

    def __init__(self, model, device: str | torch.device = "cpu",
                 use_graphs: bool = True):
        self.model = model.to(device)
        self.device = canonical_device(device)
        self.use_graphs = use_graphs and self.device.type == "cuda"
        self._graphs: dict[int, tuple] = {}
        # models with large per-row activations advertise a chunk bound
        # (e.g. the MLP's 2^20 = ~16 GiB of transient bf16 activations);
        # elementwise models batch up to 2^24 rows per graph replay
        max_bucket = int(getattr(model, "PREDICT_CHUNK", 1 << 24))
        self.buckets = [b for b in self.BUCKETS if b <= max_bucket]

    def _bucket(self, n: int) -> int:
        for b in self.buckets:
            if n <= b:
                return b
        return n

    def update_model(self, new_model) -> bool:
        """Hot-redeploy: copy new weights into the resident model's
        tensors so every captured graph stays valid (no recapture, no
        warmup forwards).  Returns False when incompatible (caller should
        build a fresh scorer)."""
        copy = getattr(self.model, "copy_weights_from", None)
        if copy is not None and copy(new_model.to(self.device)):
            return True
        return False

    def _capture(self, b: int):
        x_static = torch.zeros(b, device=self.device, dtype=torch.float32)
        # warm up kernel/launch state on a side stream before capture —
        # a SMALL batch suffices (warmup exists for lazy init, not shape)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.model.predict(x_static[: min(b, 1024)])
        torch.cuda.current_stream().wait_stream(s)
        # thread_local capture mode: the serving process has background
        # threads (uvicorn workers, torch allocator/event threads) whose
        # harmless HIP queries abort a *global*-mode capture with
        # "operation not permitted when stream is capturing" — observed
        # intermittently with 2 replicas + the runner watchdog.  One
        # retry covers transient races during capture setup itself.
        y_static = None
        for attempt in (0, 1):
            graph = torch.cuda.CUDAGraph()
            try:
                with torch.cuda.graph(graph,
                                      capture_error_mode="thread_local"):
                    y_static = self.model.predict(x_static)
                break
            except RuntimeError:
                if attempt == 1:
                    raise
                torch.cuda.synchronize()
        self._graphs[b] = (graph, x_static, y_static)
        log.info(f"captured scoring hipGraph for batch bucket {b}")

    def score_tensor(self, X: torch.Tensor) -> torch.Tensor:
        """Score a device-resident batch; returns fp32 predictions.

        Batches beyond the largest graph bucket are scored in
        largest-bucket chunks (bounds activation memory for the MLP:
        a 2^20-row chunk holds ~16 GB of bf16 activations at H=4096).
        """
        n = X.shape[0]
        if not self.use_graphs:
            return self.model.predict(X)
        max_b = self.buckets[-1]
        if n > max_b:
            out = torch.empty(n, device=self.device, dtype=torch.float32)
            for lo in range(0, n, max_b):
                hi = min(lo + max_b, n)
                out[lo:hi] = self.score_tensor(X[lo:hi])
            return out
        b = self._bucket(n)
        if b not in self._graphs:
            try:
                self._capture(b)
            except Exception as e:  # capture unavailable — direct launches
                import traceback

                log.warning(f"hipGraph capture failed for bucket {b} "
                            f"(model={type(self.model).__name__}, "
                            f"graphs={sorted(self._graphs)}): {e}; "
                            f"direct launch\n{traceback.format_exc()}")
                self.use_graphs = False
                return self.model.predict(X)
        graph, x_static, y_static = self._graphs[b]
        x_static[:n] = X.to(self.device, dtype=torch.float32)
        if n < b:
            x_static[n:] = 1.0  # benign pad (avoids log/div edge cases)
        graph.replay()
        return y_static[:n].clone()

    def score(self, X) -> np.ndarray:
        """Score a host array/list; returns fp32 numpy predictions."""
        t = torch.as_tensor(np.asarray(X, dtype=np.float32).ravel(),
                            device=self.device)
        out = self.score_tensor(t)
        return out.cpu().numpy()
