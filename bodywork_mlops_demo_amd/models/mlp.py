"""2-hidden-layer 4096-d MLP regressor — the MFMA GEMM path.

BASELINE.json config 5: "swap regressor for 2-layer MLP (4096-d, exercises
MFMA GEMM path)".  Architecture for the 1-feature regression task:

    x (n,)  --expand1d-->  h1 = relu(x * w1 + b1)      (n, H)   fused outer
    h1      --gemm_bf16-->  h2 = relu(h1 @ W2 + b2)    (n, H)   MFMA hot op
    h2      --rowdot---->   yhat = h2 @ w3 + b3        (n,)     wave-reduce

The H x H middle GEMM is the MFMA showcase; layer 1 and 3 are fused
elementwise/reduction kernels instead of degenerate K=1 / N=1 GEMMs (a
K=1 MFMA launch would waste the matrix cores — MI355X-first design).

Training: minibatch Adam on MSE; forward/backward matmuls are the
hand-written bf16 MFMA kernels (fp32 accumulate), optimizer state fp32.
In DP training gradients are bucket-all-reduced over RCCL.

Artefact: a *real sklearn ``MLPRegressor``* with ``coefs_``/``intercepts_``
injected, so ``joblib.load`` + ``predict`` work with stock sklearn
(format parity with reference ``stage_1:111-125``).
"""
from __future__ import annotations

import math

import torch

from bodywork_mlops_demo_amd.utils.device import canonical_device

from bodywork_mlops_demo_amd import ops
from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)


class GPUMLPRegressor:
    HIDDEN = 4096
    # fixed input standardisation for X ~ U(0,100) (the generator's
    # distribution, stage_3:39): mean 50, std 100/sqrt(12).  Folded into
    # the layer-1 weights of the sklearn artefact so joblib consumers see
    # raw-x semantics.
    X_MU = 50.0
    X_SIGMA = 28.86751345948129

    def __init__(self, hidden: int = 4096, device="cpu", seed: int = 7,
                 fp8_scoring: bool | None = None):
        self.hidden = hidden
        self.device = canonical_device(device)
        # opt-in MX-fp8 scoring forward (2x-rate K=128 scaled MFMA for the
        # hot h2 GEMM); training stays bf16.  Exponents are STATIC after
        # calibration so the path is hipGraph-capturable.
        if fp8_scoring is None:
            import os as _os

            fp8_scoring = _os.environ.get("BODYWORK_MLP_FP8", "0") == "1"
        self.fp8_scoring = fp8_scoring
        self._w2_q8: torch.Tensor | None = None
        self._e_w2: int | None = None
        self._e_h1: int | None = None
        g = torch.Generator(device="cpu").manual_seed(seed)
        h = hidden
        # He init, fp32 master weights
        self.w1 = torch.randn(h, generator=g) * math.sqrt(2.0)
        self.b1 = torch.zeros(h)
        self.W2 = torch.randn(h, h, generator=g) * math.sqrt(2.0 / h)
        self.b2 = torch.zeros(h)
        self.w3 = torch.randn(h, generator=g) * math.sqrt(2.0 / h)
        self.b3 = torch.zeros(1)
        self._to_device()
        self._opt_state = None

    # -- device/dtype management ------------------------------------------
    def _to_device(self):
        d = self.device
        for name in ("w1", "b1", "W2", "b2", "w3", "b3"):
            setattr(self, name, getattr(self, name).to(d))
        self._w2_q8 = None  # fp8 shadow re-created lazily on new device
        self._refresh_bf16()

    def _refresh_bf16(self):
        # W2 master is [in,out] (sklearn coefs_[1] layout); the MFMA linear
        # kernel wants K-contiguous weights both ways, so keep two bf16
        # copies: [out,in] for forward, [in,out] for backward-data.
        self.w1_bf = self.w1.bfloat16()
        self.b1_bf = self.b1.bfloat16()
        self.W2w_bf = self.W2.t().contiguous().bfloat16()   # [out,in]
        self.W2wt_bf = self.W2.contiguous().bfloat16()      # [in,out]
        self.b2_bf = self.b2.bfloat16()
        self.w3_bf = self.w3.bfloat16()

    def to(self, device):
        self.device = canonical_device(device)
        self._to_device()
        return self

    def copy_weights_from(self, other: "GPUMLPRegressor") -> bool:
        """In-place weight swap (serving hot-redeploy): masters and bf16
        shadows are copied into the EXISTING tensors so captured serving
        graphs keep valid pointers."""
        if not isinstance(other, GPUMLPRegressor) or other.hidden != self.hidden:
            return False
        for name in ("w1", "b1", "W2", "b2", "w3", "b3"):
            getattr(self, name).copy_(getattr(other, name).to(self.device))
        for name in ("w1_bf", "b1_bf", "W2w_bf", "W2wt_bf", "b2_bf", "w3_bf"):
            getattr(self, name).copy_(getattr(other, name).to(self.device))
        self._refresh_fp8()
        return True

    # -- MX-fp8 scoring shadows -------------------------------------------
    def _refresh_fp8(self) -> None:
        """Re-quantise the W2 fp8 shadow IN PLACE after a weight change.
        The shared exponent is frozen at first quantisation so captured
        graphs (which bake the int exponent into kernel args) stay valid;
        e4m3 conversion saturates gracefully if a later W2 drifts past the
        calibrated range (logged by the scorer tests, not expected for
        this model family)."""
        if self._w2_q8 is None:
            return
        self._w2_q8.copy_(ops.quantize_e4m3(self.W2w_bf, self._e_w2))

    def _ensure_fp8_weights(self) -> None:
        if self._w2_q8 is None:
            self._e_w2 = ops.e4m3_exponent(self.W2.abs().max().item())
            self._w2_q8 = ops.quantize_e4m3(self.W2w_bf, self._e_w2)
            # h1 exponent from a weight-derived bound (data-free, so it
            # can be computed in the scorer's warmup before capture):
            # h1 = relu(xn*w1 + b1) with |xn| <= (100-mu)/sigma ~= 1.74
            # for the reference's X~U(0,100) (stage_3:39).  e4m3 is a
            # floating format, so overshooting e only raises the
            # subnormal floor — it never costs relative precision.
            bound = 1.8 * self.w1.abs().max().item() \
                + self.b1.abs().max().item()
            self._e_h1 = ops.e4m3_exponent(bound)

    def parameters(self) -> list[torch.Tensor]:
        return [self.w1, self.b1, self.W2, self.b2, self.w3, self.b3]

    # -- forward -----------------------------------------------------------
    def _forward(self, x: torch.Tensor, want_masks: bool = False):
        """Forward pass; with ``want_masks`` the relu layers also emit
        their 1-bit activation masks (consumed by the masked backward —
        16x less mask traffic than re-reading the activations)."""
        xn = (x.float() - self.X_MU) / self.X_SIGMA
        if want_masks:
            h1, m1 = ops.expand1d_bf16(xn, self.w1_bf, self.b1_bf, relu=True,
                                       emit_mask=True)
            h2, m2 = ops.linear_relu_mask_bf16(h1, self.W2w_bf, self.b2_bf)
        elif self._use_fp8(xn.shape[0]):
            # fp8 scoring forward in TWO kernels total: (1) fused layer-1
            # expand emitting e4m3 directly (1 HBM byte/element), (2) the
            # 2x-rate K=128 scaled-MFMA h2 GEMM with the relu+rowdot HEAD
            # fused into its epilogue — the [M,4096] h2 activation tensor
            # never touches HBM
            self._ensure_fp8_weights()
            h1q = ops.expand1d_e4m3(xn, self.w1_bf, self.b1_bf,
                                    self._e_h1)
            yhat = ops.gemm_mx8_relu_dot(h1q, self._e_h1, self._w2_q8,
                                         self._e_w2, self.b2, self.w3)
            yhat = yhat + self.b3
            return yhat, h1q, None, xn, None, None
        elif self._use_fused_head(xn.shape[0]):
            # bf16 fused head: h2 GEMM + relu + rowdot in one kernel
            # (the fp8 path's structure at bf16 precision)
            h1 = ops.expand1d_bf16(xn, self.w1_bf, self.b1_bf, relu=True)
            yhat = ops.linear_relu_dot_bf16(h1, self.W2w_bf, self.b2,
                                            self.w3)
            yhat = yhat + self.b3
            return yhat, h1, None, xn, None, None
        else:
            h1 = ops.expand1d_bf16(xn, self.w1_bf, self.b1_bf, relu=True)
            h2 = ops.linear_bf16(h1, self.W2w_bf, bias=self.b2_bf,
                                 relu=True)
            m1 = m2 = None
        yhat = ops.rowdot_bf16(h2, self.w3_bf, self.b3)
        return yhat, h1, h2, xn, m1, m2

    def _use_fp8(self, m: int) -> bool:
        """MX-fp8 scoring is opt-in, GPU-only, and tile-shape-gated
        (the K=128 scaled-MFMA kernel serves M%256==0; other batch sizes
        take the bf16 kernel)."""
        return (self.fp8_scoring and self.device.type == "cuda"
                and m % 256 == 0 and self.hidden % 256 == 0)

    def _use_fused_head(self, m: int) -> bool:
        """bf16 fused GEMM+relu+rowdot head (same tile gating)."""
        return (self.device.type == "cuda"
                and m % 256 == 0 and self.hidden % 256 == 0)

    # Exponents on the fp8 path are STATIC: e_w2 and the weight-derived
    # e_h1 freeze at first quantisation (inside the scorer's warmup
    # forwards, before capture) — after that the whole path is
    # tensor-in/tensor-out with baked int args, so BatchedScorer can
    # hipGraph-capture it (an .item() sync would abort capture).

    #: rows per forward chunk: bounds transient activations to
    #: 2 * chunk * H bf16 (= 16 GiB at H=4096) however large the batch
    PREDICT_CHUNK = 1 << 20

    def predict(self, X: torch.Tensor) -> torch.Tensor:
        X = X.to(self.device)
        n = X.shape[0]
        if n <= self.PREDICT_CHUNK:
            return self._forward(X)[0]
        out = torch.empty(n, dtype=torch.float32, device=self.device)
        for lo in range(0, n, self.PREDICT_CHUNK):
            hi = min(lo + self.PREDICT_CHUNK, n)
            out[lo:hi] = self._forward(X[lo:hi])[0]
        return out

    # -- training ----------------------------------------------------------
    def fit(
        self,
        X: torch.Tensor,
        y: torch.Tensor,
        steps: int = 200,
        batch_size: int = 65536,
        lr: float = 3e-4,
        process_group=None,
        seed: int = 42,
    ):
        """Minibatch Adam on MSE.  With ``process_group``, each rank holds
        a shard of (X, y); gradients are all-reduce-averaged per step
        (flat fp32 bucket — one RCCL launch per step)."""
        n = X.shape[0]
        g = torch.Generator(device="cpu").manual_seed(seed)
        if self._opt_state is None:
            self._opt_state = [
                (torch.zeros_like(p), torch.zeros_like(p)) for p in self.parameters()
            ]
        t = 0
        world = 1
        if process_group is not None:
            import torch.distributed as dist

            world = dist.get_world_size(process_group)

        # graph path: the ENTIRE Adam step (philox batch sampling,
        # gather, forward, backward, [DP: RCCL all-reduce], fused Adam +
        # shadow refresh) is one captured hipGraph, replayed per step —
        # host dispatch of the ~30 kernels was costing as much as the
        # kernels themselves.  With a process_group the all-reduce is
        # captured too (RCCL collectives are hipGraph-capturable); if
        # capture fails on this build we fall back to the eager DP loop.
        graph_ok = (self.device.type == "cuda" and steps >= 4
                    and ops.hip_available()
                    and not getattr(self, "_dp_graph_unsupported", False))
        if graph_ok and process_group is not None:
            import torch.distributed as dist

            graph_ok = dist.get_backend(process_group) == "nccl"
        if graph_ok:
            try:
                self._fit_captured(X.to(self.device), y.to(self.device),
                                   steps, min(batch_size, n), lr, seed,
                                   process_group)
                return self
            except RuntimeError:
                if process_group is None:
                    raise
                log.warning("hipGraph capture of the DP step failed; "
                            "falling back to the eager per-step loop")
                self._dp_graph_unsupported = True
                if getattr(self, "_train_static", None) is not None:
                    self._train_static["graph"] = None

        for _ in range(steps):
            idx = torch.randint(0, n, (min(batch_size, n),), generator=g)
            xb = X[idx.to(X.device)]
            yb = y[idx.to(y.device)]
            grads = self._step_grads(xb, yb)
            if process_group is not None:
                import torch.distributed as dist

                flat = torch.cat([gr.reshape(-1) for gr in grads])
                dist.all_reduce(flat, group=process_group)
                flat /= world
                off = 0
                for i, gr in enumerate(grads):
                    grads[i] = flat[off:off + gr.numel()].view_as(gr)
                    off += gr.numel()
            t += 1
            self._adam_update(grads, lr, t)
        return self

    def reinit_(self, seed: int = 7) -> "GPUMLPRegressor":
        """Re-randomise weights and reset optimiser state IN PLACE, so a
        cached training graph (whose nodes hold these tensor pointers)
        stays valid for the next day's fresh fit.

        Draws come from the same CPU generator stream, in the same order
        and pattern, as ``__init__`` — so ``reinit_(s)`` on a warm cache
        and a cold ``GPUMLPRegressor(seed=s)`` start from bit-identical
        weights, keeping results reproducible across resume boundaries."""
        g = torch.Generator(device="cpu").manual_seed(seed)
        h = self.hidden
        self.w1.copy_(torch.randn(h, generator=g) * math.sqrt(2.0))
        self.b1.zero_()
        self.W2.copy_(torch.randn(h, h, generator=g) * math.sqrt(2.0 / h))
        self.b2.zero_()
        self.w3.copy_(torch.randn(h, generator=g) * math.sqrt(2.0 / h))
        self.b3.zero_()
        if self._opt_state is not None:
            for m, v in self._opt_state:
                m.zero_()
                v.zero_()
        # refresh shadows in place (pointer-stable)
        self.w1_bf.copy_(self.w1.bfloat16())
        self.b1_bf.copy_(self.b1.bfloat16())
        self.W2wt_bf.copy_(self.W2.bfloat16())
        self.W2w_bf.copy_(ops.transpose_bf16(self.W2wt_bf)
                          if self.device.type == "cuda"
                          else self.W2.t().contiguous().bfloat16())
        self.b2_bf.copy_(self.b2.bfloat16())
        self.w3_bf.copy_(self.w3.bfloat16())
        self._refresh_fp8()
        return self

    def _fit_captured(self, X, y, steps: int, bs: int, lr: float, seed: int,
                      process_group=None):
        """hipGraph-captured Adam steps (see fit()).  Bias correction is a
        device [2]-tensor the graph reads; the philox batch counter and
        data size live on-device; the day's data is copied into persistent
        capacity buffers — so ONE captured graph serves every retrain of
        the drift loop (capture cost paid once, replay ~zero host cost)."""
        n = X.shape[0]
        beta1, beta2 = 0.9, 0.999
        world = 1
        if process_group is not None:
            import torch.distributed as dist

            world = dist.get_world_size(process_group)
        st = getattr(self, "_train_static", None)
        if (st is None or st["cap"] < n or st["bs"] != bs or st["lr"] != lr
                or st.get("world", 1) != world):
            cap = max(n, int(st["cap"]) if st else 0)
            st = {
                "cap": cap, "bs": bs, "lr": lr,
                "X": torch.empty(cap, device=self.device),
                "y": torch.empty(cap, device=self.device),
                "n_dev": torch.zeros(1, dtype=torch.int64, device=self.device),
                "ctr": torch.zeros(1, dtype=torch.int64, device=self.device),
                "bc": torch.ones(2, device=self.device),
                "graph": None, "world": world,
            }
            self._train_static = st
        st["X"][:n].copy_(X)
        st["y"][:n].copy_(y)
        st["n_dev"].fill_(n)
        st["ctr"].zero_()
        bc = st["bc"]

        def one_step():
            idx = ops.batch_indices(st["ctr"], st["n_dev"], bs, seed)
            xb = st["X"].index_select(0, idx)
            yb = st["y"].index_select(0, idx)
            grads = self._step_grads(xb, yb)
            if process_group is not None:
                import torch.distributed as dist

                # one flat fp32 bucket -> one RCCL launch per step, same
                # as the eager DP loop; allocations made during capture
                # come from the graph pool and are stable across replays
                flat = torch.cat([gr.reshape(-1) for gr in grads])
                dist.all_reduce(flat, group=process_group)
                flat /= world
                off = 0
                for i, gr in enumerate(grads):
                    grads[i] = flat[off:off + gr.numel()].view_as(gr)
                    off += gr.numel()
            shadows = [self.w1_bf, self.b1_bf, self.W2wt_bf, self.b2_bf,
                       self.w3_bf, None]
            for p, gr, (m, v), sh in zip(self.parameters(), grads,
                                         self._opt_state, shadows):
                ops.adam_step(p, gr, m, v, sh, lr, t=1, bc=bc)
            # refresh the transposed forward copy in place (graph-safe)
            self.W2w_bf.copy_(ops.transpose_bf16(self.W2wt_bf))

        t0 = 1
        if st["graph"] is None:
            # warmup on a side stream (this executes one real step), then
            # capture (recording only — not an executed step)
            stream = torch.cuda.Stream()
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                bc.copy_(torch.tensor([1 / (1 - beta1), 1 / (1 - beta2)]))
                one_step()
            torch.cuda.current_stream().wait_stream(stream)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                one_step()
            st["graph"] = graph
            t0 = 2  # the warmup was step 1
        graph = st["graph"]
        for t in range(t0, steps + 1):
            bc.copy_(torch.tensor([1 / (1 - beta1**t), 1 / (1 - beta2**t)],
                                  dtype=torch.float32), non_blocking=True)
            graph.replay()

    def _step_grads(self, xb: torch.Tensor, yb: torch.Tensor) -> list[torch.Tensor]:
        nb = xb.shape[0]
        yhat, h1, h2, xn, m1, m2 = self._forward(xb, want_masks=True)
        dy = (2.0 / nb) * (yhat - yb.float())                   # (n,)
        # layer 3: yhat = h2 @ w3 + b3
        dw3 = ops.coldot_bf16(h2, dy)                           # (H,)
        db3 = dy.sum().reshape(1)
        # dh2 = outer(dy, w3) * relu'(h2): fused expand with 1-bit mask
        dz2 = ops.expand1d_bf16(dy, self.w3_bf, None, relu=False, mask=m2)
        # layer 2: h2 = relu(h1 @ W2 + b2), W2 [in,out]
        dW2 = ops.gemm_tn_bf16(h1, dz2, out_fp32=True)          # (in,out)
        db2 = ops.colsum_bf16(dz2)
        dz1 = ops.linear_bf16(dz2, self.W2wt_bf, mask=m1)       # dz2 @ W2^T
        # layer 1: h1 = relu(xn w1 + b1)
        dw1, db1 = ops.coldot_bf16(dz1, xn, also_colsum=True)
        return [dw1, db1, dW2, db2, dw3, db3]

    def _adam_update(self, grads, lr: float, t: int, beta1=0.9, beta2=0.999, eps=1e-8):
        """Fused HIP Adam on GPU (updates master fp32 + bf16 shadow in one
        pass per parameter; one tiled-transpose for the [out,in] copy);
        plain torch on CPU."""
        if self.device.type == "cuda":
            shadows = [self.w1_bf, self.b1_bf, self.W2wt_bf, self.b2_bf,
                       self.w3_bf, None]
            for p, gr, (m, v), s in zip(self.parameters(), grads,
                                        self._opt_state, shadows):
                ops.adam_step(p, gr, m, v, s, lr, t, beta1, beta2, eps)
            self.W2w_bf = ops.transpose_to_bf16(self.W2)
            return
        for p, gr, (m, v) in zip(self.parameters(), grads, self._opt_state):
            ops.adam_step(p, gr, m, v, None, lr, t, beta1, beta2, eps)
        self._refresh_bf16()

    # -- artefact compatibility --------------------------------------------
    def to_sklearn(self):
        """Real sklearn MLPRegressor artefact (raw-x semantics: the input
        standardisation is folded into layer 1, so stock sklearn
        ``predict`` on raw X matches our forward).  fp32 arrays keep the
        4096-d artefact at ~67 MB instead of 134."""
        import numpy as np
        from sklearn.neural_network import MLPRegressor

        h = self.hidden
        m = MLPRegressor(hidden_layer_sizes=(h, h), activation="relu")
        w1 = self.w1.detach().cpu().numpy()
        b1 = self.b1.detach().cpu().numpy()
        m.coefs_ = [
            (w1 / self.X_SIGMA).reshape(1, h).astype(np.float32),
            self.W2.detach().cpu().numpy().astype(np.float32),
            self.w3.detach().cpu().numpy().reshape(h, 1).astype(np.float32),
        ]
        m.intercepts_ = [
            (b1 - w1 * (self.X_MU / self.X_SIGMA)).astype(np.float32),
            self.b2.detach().cpu().numpy().astype(np.float32),
            self.b3.detach().cpu().numpy().astype(np.float32),
        ]
        m.n_layers_ = 4
        m.n_outputs_ = 1
        m.out_activation_ = "identity"
        m.n_features_in_ = 1
        return m

    @classmethod
    def from_sklearn(cls, m, device="cpu") -> "GPUMLPRegressor":
        h = m.coefs_[1].shape[0]
        self = cls.__new__(cls)
        self.hidden = h
        self.device = canonical_device(device)
        wr = torch.from_numpy(m.coefs_[0].reshape(-1).copy()).float()
        br = torch.from_numpy(m.intercepts_[0].copy()).float()
        # invert the raw-x fold: wr = w1/sigma, br = b1 - wr*mu
        self.w1 = wr * cls.X_SIGMA
        self.b1 = br + wr * cls.X_MU
        self.W2 = torch.from_numpy(m.coefs_[1].copy()).float()
        self.b2 = torch.from_numpy(m.intercepts_[1].copy()).float()
        self.w3 = torch.from_numpy(m.coefs_[2].reshape(-1).copy()).float()
        self.b3 = torch.from_numpy(m.intercepts_[2].copy()).float()
        self._opt_state = None
        import os as _os

        self.fp8_scoring = _os.environ.get("BODYWORK_MLP_FP8", "0") == "1"
        self._w2_q8 = None
        self._e_w2 = None
        self._e_h1 = None
        self._to_device()
        return self

    def __repr__(self) -> str:
        return f"MLPRegressor(hidden_layer_sizes=({self.hidden}, {self.hidden}))"
