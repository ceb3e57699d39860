"""Server-side serving throughput over the binary wire, per model family.

Trains a model into a fresh store, starts ONE uvicorn replica, streams
`--batch`-row float32 batches at `POST /score/v1/binary` and prints
per-request and steady-state rows/s plus the server's own /stats.  The
round-1 linear measurement (129-189M rows/s) came from an inline script;
this makes it reproducible and adds the MLP bf16/fp8 variants.

    python tools/serve_throughput.py --model linear|mlp|mlp-fp8
"""
from __future__ import annotations

import argparse
import os
import subprocess
import sys
import tempfile
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="linear",
                   choices=["linear", "mlp", "mlp-fp8"])
    p.add_argument("--rows", type=int, default=200_000,
                   help="training rows")
    p.add_argument("--batch", type=int, default=4_000_000)
    p.add_argument("--requests", type=int, default=8)
    p.add_argument("--port", type=int, default=5721)
    p.add_argument("--mlp-steps", type=int, default=10)
    args = p.parse_args()

    if args.model == "mlp-fp8":
        os.environ["BODYWORK_MLP_FP8"] = "1"
        args.model = "mlp"
        tag = "mlp-fp8"
    else:
        tag = args.model

    import requests
    import torch

    from bodywork_mlops_demo_amd.stages import datagen, train
    from bodywork_mlops_demo_amd.store import LocalStore

    device = "cuda" if torch.cuda.is_available() else "cpu"
    store_dir = tempfile.mkdtemp(prefix="serve-tp-")
    store = LocalStore(store_dir)
    from datetime import date

    datagen.run(store, n=args.rows, date=date(2026, 6, 1), device=device)
    train.run(store, model_type=args.model, device=device,
              mlp_steps=args.mlp_steps)

    proc = subprocess.Popen(
        [sys.executable, "-m", "bodywork_mlops_demo_amd.stages.serve",
         "--store", store_dir, "--host", "127.0.0.1",
         "--port", str(args.port)], env=dict(os.environ))
    url = f"http://127.0.0.1:{args.port}"
    try:
        deadline = time.time() + 240
        while time.time() < deadline:
            try:
                if requests.get(f"{url}/healthz", timeout=2).json().get(
                        "status") == "ok":
                    break
            except Exception:
                time.sleep(0.25)
        else:
            raise RuntimeError("server never became healthy")

        rng = np.random.default_rng(0)
        X = rng.uniform(0, 100, args.batch).astype(np.float32)
        body = X.tobytes()
        print(f"=== {tag}: binary wire, {args.batch}-row batches ===",
              flush=True)
        rates = []
        session = requests.Session()
        for i in range(args.requests):
            t0 = time.perf_counter()
            r = session.post(f"{url}/score/v1/binary", data=body,
                             headers={"Content-Type":
                                      "application/octet-stream"},
                             timeout=300)
            dt = time.perf_counter() - t0
            assert r.ok, r.status_code
            n = len(r.content) // 4
            rates.append(n / dt)
            print(f"req {i}: {n} rows in {dt * 1e3:.1f} ms = "
                  f"{n / dt / 1e6:.1f} M rows/s", flush=True)
        steady = sorted(rates[2:])[len(rates[2:]) // 2] if len(rates) > 4 \
            else max(rates)
        stats = session.get(f"{url}/stats", timeout=5).json()
        print(f"steady-state median: {steady / 1e6:.1f} M rows/s; "
              f"server p99 {stats['p99_s'] * 1e3:.2f} ms, "
              f"server-side {stats['rows_per_sec'] / 1e6:.1f} M rows/s")
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except Exception:
            proc.kill()


if __name__ == "__main__":
    main()
