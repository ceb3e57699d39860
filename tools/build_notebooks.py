"""Build the 5 notebook mirrors WITH executed outputs (reference C9).

The reference notebooks (~1,490 lines of JSON) carry executed outputs,
the alpha(d) drift plot and a curl walkthrough (``notebooks/README.md``,
``3-generate-next-dataset.ipynb``, ``stage_2_serve_model.py:8-22``).
This image has no jupyter/nbconvert, so this builder executes each code
cell in-process (shared namespace per notebook, stdout + last-expression
+ matplotlib figures captured) and writes standard .ipynb JSON that any
Jupyter renders — the committed notebooks are real executed documents,
and re-running this script regenerates them from scratch against a fresh
seeded store:

    python tools/build_notebooks.py [--store DIR] [--out notebooks/]
"""
from __future__ import annotations

import argparse
import ast
import base64
import contextlib
import io
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import matplotlib

matplotlib.use("Agg")


class Notebook:
    def __init__(self, ns: dict):
        self.cells: list[dict] = []
        self.ns = ns
        self.count = 0

    def md(self, text: str) -> None:
        self.cells.append({
            "cell_type": "markdown", "metadata": {},
            "source": text.strip("\n").splitlines(keepends=True),
        })

    def code(self, src: str) -> None:
        import matplotlib.pyplot as plt

        src = src.strip("\n")
        self.count += 1
        outputs: list[dict] = []
        buf = io.StringIO()
        tree = ast.parse(src)
        # notebook semantics: echo the value of a trailing bare expression
        last_expr = None
        if tree.body and isinstance(tree.body[-1], ast.Expr):
            last_expr = ast.Expression(tree.body.pop().value)
        with contextlib.redirect_stdout(buf):
            exec(compile(tree, "<cell>", "exec"), self.ns)
            result = (eval(compile(last_expr, "<cell>", "eval"), self.ns)
                      if last_expr is not None else None)
        text = buf.getvalue()
        if text:
            outputs.append({"output_type": "stream", "name": "stdout",
                            "text": text.splitlines(keepends=True)})
        for num in plt.get_fignums():
            fig = plt.figure(num)
            png = io.BytesIO()
            fig.savefig(png, format="png", dpi=80, bbox_inches="tight")
            outputs.append({
                "output_type": "display_data",
                "data": {"image/png":
                         base64.b64encode(png.getvalue()).decode()},
                "metadata": {},
            })
        plt.close("all")
        if result is not None:
            outputs.append({
                "output_type": "execute_result",
                "execution_count": self.count,
                "data": {"text/plain": repr(result).splitlines(keepends=True)},
                "metadata": {},
            })
        self.cells.append({
            "cell_type": "code", "execution_count": self.count,
            "metadata": {}, "outputs": outputs,
            "source": src.splitlines(keepends=True),
        })

    def save(self, path: str) -> None:
        nb = {
            "cells": self.cells,
            "metadata": {
                "kernelspec": {"display_name": "Python 3",
                               "language": "python", "name": "python3"},
                "language_info": {"name": "python",
                                  "version": sys.version.split()[0]},
            },
            "nbformat": 4, "nbformat_minor": 5,
        }
        with open(path, "w") as f:
            json.dump(nb, f, indent=1)
        print(f"wrote {path} ({len(self.cells)} cells, "
              f"{os.path.getsize(path)} bytes)")


STORE_CELL = """
import os
import torch
from bodywork_mlops_demo_amd.store import open_store

# BODYWORK_AMD_STORE points at the pipeline's artefact store
# (../artefact-store when running next to a local pipeline run)
store = open_store(os.environ.get("BODYWORK_AMD_STORE", "../artefact-store"))
device = "cuda" if torch.cuda.is_available() else "cpu"
store, device
"""


def build_train(ns) -> Notebook:
    nb = Notebook(ns)
    nb.md("""
# Train Model (stage 1, interactive)

Interactive mirror of `stage-1-train-model` (reference
`notebooks/1-train-model.ipynb`): load **all** accumulated datasets from
the artefact store in time order, fit the regressor — on MI355X the OLS
fit is one fused XᵀX/Xᵀy HIP statistics reduction
(`ops/hip/linreg.hip`), not an sklearn lstsq — compute the offline
metrics (MAPE / R² / max residual, reference `stage_1:79-90`), and
persist the **joblib** model + metrics CSV keyed by the newest dataset
date.
""")
    nb.code(STORE_CELL)
    nb.md("Load the full dataset history (reference `stage_1:59-71` "
          "read-all semantics):")
    nb.code("""
y, X, latest = store.get_all_datasets()
print(f"{len(y)} rows through {latest}")
""")
    nb.md("Train + persist through the stage-1 library entry (the same "
          "code the pipeline runs):")
    nb.code("""
from bodywork_mlops_demo_amd.stages import train

metrics = train.run(store, model_type="linear", device=device)
metrics
""")
    nb.md("The artefact store now holds the dated joblib model and its "
          "offline metrics (the dated key IS the version — reference "
          "`stage_1:113,130`):")
    nb.code("""
store.list_keys("models/")[-3:] + store.list_keys("model-metrics/")[-3:]
""")
    return nb


def build_serve(ns) -> Notebook:
    nb = Notebook(ns)
    nb.md("""
# Serve Model (stage 2, interactive)

Interactive mirror of `stage-2-serve-model`: start a serving replica
(uvicorn, model resident in device memory, hipGraph-captured batch
scoring on GPU), then exercise the wire-compatible scoring API
(reference `stage_2_serve_model.py:73-80`).
""")
    nb.code(STORE_CELL)
    nb.code("""
import subprocess, sys, time, requests

PORT = 5209
proc = subprocess.Popen(
    [sys.executable, "-m", "bodywork_mlops_demo_amd.stages.serve",
     "--store", store.uri, "--host", "127.0.0.1", "--port", str(PORT)])
for _ in range(120):
    try:
        if requests.get(f"http://127.0.0.1:{PORT}/healthz",
                        timeout=2).json().get("status") == "ok":
            break
    except Exception:
        time.sleep(0.25)
requests.get(f"http://127.0.0.1:{PORT}/healthz", timeout=5).json()
""")
    nb.md("""
Score a single instance — the reference wire format
(`{"X": x}` → `{"prediction": p, "model_info": s}`,
`stage_2:11-21`):
""")
    nb.code("""
r = requests.post(f"http://127.0.0.1:{PORT}/score/v1",
                  json={"X": 50}, timeout=10)
r.json()
""")
    nb.md("""
The same request from a shell (the reference docstring's manual check,
`stage_2_serve_model.py:8-22`):

```sh
curl http://127.0.0.1:5209/score/v1 \\
    --request POST \\
    --header "Content-Type: application/json" \\
    --data '{"X": 50}'
```

Expected response shape:

```json
{"prediction": <float>, "model_info": "LinearRegression()"}
```

Batch extensions beyond the reference: `POST /score/v1/batch` scores a
list in one fused GPU launch, and `POST /score/v1/binary` takes raw
float32 (the high-throughput wire — 129-189 M rows/s measured on one
MI355X, `profiles/r01_serving_throughput_final.log`).
""")
    nb.code("""
batch = requests.post(f"http://127.0.0.1:{PORT}/score/v1/batch",
                      json={"X": [0.0, 25.0, 50.0, 75.0, 100.0]},
                      timeout=10).json()
stats = requests.get(f"http://127.0.0.1:{PORT}/stats", timeout=5).json()
proc.terminate(); proc.wait(timeout=10)
batch, stats
""")
    return nb


def build_datagen(ns) -> Notebook:
    nb = Notebook(ns)
    nb.md(r"""
# Generate Next Dataset (stage 3, interactive)

Interactive mirror of `stage-3-generate-next-dataset`, with the
generator's maths (reference `3-generate-next-dataset.ipynb`).  The
daily dataset is

$$ y = \alpha(d) + \beta X + \sigma \epsilon,\qquad
   X \sim U(0, 100),\quad \epsilon \sim N(0, 1) $$

with $\beta = 0.5$, $\sigma = 10$, rows with $y < 0$ culled, and the
**concept drift** carried by the sinusoidal intercept over the day of
the year $d$:

$$ \alpha(d) = \kappa + A \sin\!\left(2\pi f \frac{d-1}{364}\right),
   \qquad f = 6,\ \kappa = 1,\ A = 0.5 $$

(reference `stage_3_synthetic_data_generation.py:31-41`).  On MI355X the
whole generator runs on-GPU: philox4x32 uniform/Box-Muller draws and a
stable in-kernel $y \ge 0$ stream compaction
(`ops/hip/datagen.hip`).
""")
    nb.code(STORE_CELL)
    nb.md("The drift knob: $\\alpha(d)$ over a full year —")
    nb.code("""
import numpy as np
import matplotlib.pyplot as plt

d = np.arange(1, 366)
alpha = 1.0 + 0.5 * np.sin(2 * np.pi * 6 * (d - 1) / 364)
plt.figure(figsize=(9, 3))
plt.plot(d, alpha)
plt.xlabel("day of year d"); plt.ylabel(r"$\\alpha(d)$")
plt.title(r"sinusoidal concept drift: $\\alpha(d)$, f=6, $\\kappa$=1, A=0.5")
plt.grid(alpha=0.3)
""")
    nb.md("Generate and persist the next day's dataset (dated key = "
          "version, reference `stage_3:49`):")
    nb.code("""
from datetime import timedelta
from bodywork_mlops_demo_amd.stages import datagen
from bodywork_mlops_demo_amd.store import contract

_, latest = store.latest(contract.DATASETS_PREFIX)
next_day = latest + timedelta(days=1)
key = datagen.run(store, n=1440, date=next_day, device=device)
key
""")
    nb.md("The generated joint distribution:")
    nb.code("""
y, X = store.get_dataset(contract.dataset_key(next_day))
plt.figure(figsize=(5, 4))
plt.scatter(X, y, s=4, alpha=0.4)
plt.xlabel("X"); plt.ylabel("y")
plt.title(f"regression-dataset-{next_day} ({len(y)} rows, y>=0 culled)")
""")
    return nb


def build_loadtest(ns) -> Notebook:
    nb = Notebook(ns)
    nb.md("""
# Test Model Scoring Service (stage 4, interactive)

Interactive mirror of `stage-4-test-model-scoring-service`: score the
latest (t+1) dataset against the LIVE service — so the model trained on
data through period *t* is always evaluated on unseen next-period data,
which is what makes drift visible (reference
`4-test-model-scoring-service.ipynb` intro) — then persist the online
test metrics (MAPE, score-label correlation, max APE, mean response
time; reference `stage_4:101-113`).
""")
    nb.code(STORE_CELL)
    nb.code("""
import subprocess, sys, time, requests

PORT = 5209
proc = subprocess.Popen(
    [sys.executable, "-m", "bodywork_mlops_demo_amd.stages.serve",
     "--store", store.uri, "--host", "127.0.0.1", "--port", str(PORT)])
for _ in range(120):
    try:
        if requests.get(f"http://127.0.0.1:{PORT}/healthz",
                        timeout=2).json().get("status") == "ok":
            break
    except Exception:
        time.sleep(0.25)
""")
    nb.md("Drive the load test through the stage-4 library entry "
          "(`mode='batch'` = chunked requests, one fused GPU launch per "
          "chunk; `mode='serial'` reproduces the reference's "
          "one-POST-per-row client, `stage_4:68-85`):")
    nb.code("""
from bodywork_mlops_demo_amd.stages import loadtest

metrics = loadtest.run(store, url=f"http://127.0.0.1:{PORT}/score/v1",
                       mode="batch", device=device)
proc.terminate(); proc.wait(timeout=10)
metrics
""")
    nb.md("The durable test-metrics artefact (reference `stage_4:122`):")
    nb.code("""
from bodywork_mlops_demo_amd.store import contract

keys = store.list_keys(contract.TEST_METRICS_PREFIX)
store.get_metrics_csv(keys[-1])
""")
    return nb


def build_analytics(ns) -> Notebook:
    nb = Notebook(ns)
    nb.md("""
# Model Performance Analytics

Mirror of the reference's `model-performance-analytics.ipynb`: download
**all** historical `model-metrics/` and `test-metrics/` CSVs, concat
them into two time-ordered DataFrames, and inspect offline-vs-online
divergence — the human-in-the-loop drift regression check.
""")
    nb.code(STORE_CELL)
    nb.code("""
from bodywork_mlops_demo_amd.monitoring.analytics import (
    download_metrics, drift_report)
from bodywork_mlops_demo_amd.store import contract

offline = download_metrics(store, contract.MODEL_METRICS_PREFIX)
online = download_metrics(store, contract.TEST_METRICS_PREFIX)
offline.tail()
""")
    nb.code("""
report = drift_report(store)
report["summary"]
""")
    nb.md("Offline vs online MAPE over the simulated days — the gap "
          "widening between retrains is the concept drift:")
    nb.code("""
import matplotlib.pyplot as plt

j = report["joined"]
plt.figure(figsize=(9, 3.5))
plt.plot(j["date"], j["MAPE_offline"], "o-", label="offline MAPE (train-time)")
plt.plot(j["date"], j["MAPE_online"], "s-", label="online MAPE (live service, t+1 data)")
plt.legend(); plt.grid(alpha=0.3); plt.xticks(rotation=30)
plt.title("offline vs online model quality across pipeline days")
""")
    nb.code("""
plt.figure(figsize=(9, 3))
plt.plot(j["date"], j["MAPE_online"] - j["MAPE_offline"], "k.-")
plt.axhline(0, color="gray", lw=0.5)
plt.grid(alpha=0.3); plt.xticks(rotation=30)
plt.title("drift gap: online - offline MAPE")
""")
    return nb


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--store", default=None,
                   help="existing artefact store (default: fresh tmp store "
                        "seeded with a 6-day drift loop)")
    p.add_argument("--out", default="notebooks")
    args = p.parse_args()

    if args.store is None:
        import tempfile

        store_dir = tempfile.mkdtemp(prefix="bodywork-nb-store-")
        from bodywork_mlops_demo_amd.pipeline.loop import run_loop
        from bodywork_mlops_demo_amd.store import LocalStore

        print(f"seeding {store_dir} with a 6-day drift loop ...")
        run_loop(LocalStore(store_dir), days=6, n_rows=1440,
                 model_type="linear", device="cpu",
                 start_date="2026-01-01")
        args.store = store_dir
    os.environ["BODYWORK_AMD_STORE"] = args.store

    builders = {
        "1-train-model.ipynb": build_train,
        "2-serve-model.ipynb": build_serve,
        "3-generate-next-dataset.ipynb": build_datagen,
        "4-test-model-scoring-service.ipynb": build_loadtest,
        "model-performance-analytics.ipynb": build_analytics,
    }
    os.makedirs(args.out, exist_ok=True)
    for name, builder in builders.items():
        nb = builder({})
        nb.save(os.path.join(args.out, name))


if __name__ == "__main__":
    main()
