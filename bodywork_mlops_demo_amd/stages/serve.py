"""stage-2-serve-model: the model-scoring REST service.

Reference semantics (``stage_2_serve_model.py:108-119``): at startup,
download the latest model by key-date and hold it in a process global;
serve ``POST /score/v1`` on port 5000.

MI355X path: one replica process per GPU (the pipeline runner fans out
``service.replicas`` processes, pinning ``HIP_VISIBLE_DEVICES`` — the
shared-nothing replication of ``bodywork.yaml:40``), model resident in
HBM, batched hipGraph scoring (:mod:`bodywork_mlops_demo_amd.serving`).
"""
from __future__ import annotations

import argparse
import os

from bodywork_mlops_demo_amd.monitoring import stage_guard
from bodywork_mlops_demo_amd.store import open_store
from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)

STAGE_NAME = "stage-2-serve-model"


def main(argv=None) -> None:
    import torch
    import uvicorn

    from bodywork_mlops_demo_amd.serving.server import create_app

    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--store", default=None)
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=int(os.environ.get("PORT", 5000)))
    p.add_argument("--device", default=None)
    p.add_argument("--no-graphs", action="store_true")
    args = p.parse_args(argv)

    device = args.device
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"

    with stage_guard(STAGE_NAME, exit_on_error=True):
        app = create_app(open_store(args.store), device=device,
                         use_graphs=not args.no_graphs)
        log.info(f"starting API server on {args.host}:{args.port} ({device})")
        uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
