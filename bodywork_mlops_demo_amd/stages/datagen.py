"""stage-3-generate-next-dataset: synthetic concept-drift data generator.

Reference semantics (``stage_3_synthetic_data_generation.py:28-43``):
``y = alpha(d) + 0.5*X + 10*eps`` with ``X ~ U(0,100)``, ``eps ~ N(0,1)``,
``alpha(d) = 1 + 0.5*sin(2*pi*6*(d-1)/364)``, rows with ``y < 0`` culled,
N = 24*60 rows per day, persisted as a dated CSV.

MI355X path: the whole generator (philox RNG, drift model, y>=0 stream
compaction) is one HIP kernel pass producing device-resident columns; the
CSV is written only at the artefact boundary (binary ``.npz`` format
available for large N — SURVEY.md §7 hard part (e)).
"""
from __future__ import annotations

import argparse
from datetime import date as date_t

import torch

from bodywork_mlops_demo_amd import ops
from bodywork_mlops_demo_amd.monitoring import stage_guard
from bodywork_mlops_demo_amd.store import ArtefactStore, open_store
from bodywork_mlops_demo_amd.utils.clock import CLOCK
from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)

STAGE_NAME = "stage-3-generate-next-dataset"
N_DEFAULT = 24 * 60  # reference stage_3:19


def run(
    store: ArtefactStore,
    n: int = N_DEFAULT,
    date: date_t | None = None,
    device: str | None = None,
    seed: int | None = None,
    fmt: str = "csv",
    persist: bool = True,
) -> tuple[torch.Tensor, torch.Tensor, date_t]:
    """Generate + persist one day's dataset; returns device-resident (y, X)."""
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    date = date or CLOCK.today()
    # deterministic per-day stream unless a seed is given
    seed = seed if seed is not None else date.toordinal()
    y, X = ops.datagen(n, date.timetuple().tm_yday, seed, device=device)
    log.info(
        f"generated {y.shape[0]}/{n} rows for {date} "
        f"(alpha={ops.alpha(date.timetuple().tm_yday):.4f}, device={device})"
    )
    if persist:
        key = store.put_dataset(date, y.cpu().numpy(), X.cpu().numpy(), fmt=fmt)
        log.info(f"uploaded dataset to {key}")
    return y, X, date


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--store", default=None)
    p.add_argument("--n", type=int, default=N_DEFAULT)
    p.add_argument("--date", default=None, help="YYYY-MM-DD (default: clock)")
    p.add_argument("--device", default=None)
    p.add_argument("--format", default="csv", choices=["csv", "npy"])
    args = p.parse_args(argv)
    d = date_t.fromisoformat(args.date) if args.date else None
    with stage_guard(STAGE_NAME, exit_on_error=True):
        run(open_store(args.store), n=args.n, date=d, device=args.device,
            fmt=args.format)


if __name__ == "__main__":
    main()
