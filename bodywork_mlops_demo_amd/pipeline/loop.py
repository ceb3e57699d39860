"""Multi-day concept-drift loop (BASELINE config 4).

The reference runs one cycle per *real* day via a k8s cronjob
(``README.md:5``); its stages stamp artefacts with ``date.today()`` so
more than one cycle/day is impossible.  This loop
parameterises the date (virtual clock) and runs N retrain-and-redeploy
cycles back-to-back, reporting the per-cycle wall-clock and leaving the
full dated artefact history in the store — which is also how
checkpoint/resume works: the loop resumes from the latest dataset date
found in the store (the artefacts ARE the checkpoints).
"""
from __future__ import annotations

import argparse
import json
from datetime import date as date_t

import torch

from bodywork_mlops_demo_amd.pipeline.cycle import CycleState, run_cycle
from bodywork_mlops_demo_amd.store import ArtefactStore, contract, open_store
from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)


def run_loop(
    store: ArtefactStore | None,
    days: int = 30,
    n_rows: int = 24 * 60,
    model_type: str = "linear",
    device: str | None = None,
    start_date: date_t | str = "2026-01-01",
    persist_fmt: str = "csv",
    process_group=None,
    rank: int = 0,
    world_size: int = 1,
    resume: bool = True,
    retrain_policy: str = "always",
    drift_threshold: float = 1.5,
) -> list[dict]:
    """Run ``days`` cycles.

    ``retrain_policy``:
    - ``"always"`` — reference semantics: retrain + redeploy every day.
    - ``"drift"``  — monitoring-driven MLOps: keep the deployed model
      while its live MAPE stays below ``drift_threshold`` x its own
      offline MAPE at training time; retrain only when the online
      metrics show the concept has drifted away from it.
    """
    device = device or ("cuda" if torch.cuda.is_available() else "cpu")
    if isinstance(start_date, str):
        start_date = date_t.fromisoformat(start_date)

    # resume from the latest dataset date in the store (artefact=checkpoint)
    if store is not None and resume:
        try:
            _, latest = store.latest(contract.DATASETS_PREFIX)
            start_date = latest
            log.info(f"resuming drift loop from latest store date {latest}")
        except FileNotFoundError:
            pass

    state = CycleState(device, start_date, rank=rank, world_size=world_size)

    # preload history from the store so training sees all prior days
    if store is not None:
        try:
            y_np, X_np, _ = store.get_all_datasets()
            if world_size > 1:
                # cycles generate per-rank shards; resuming from the
                # store must re-shard the merged history the same way
                # (train.py's rank::world convention) or every rank
                # would train on duplicated data
                y_np = y_np[rank::world_size].copy()
                X_np = X_np[rank::world_size].copy()
            state.append_day(
                torch.from_numpy(y_np).to(device),
                torch.from_numpy(X_np).to(device),
            )
            log.info(f"preloaded {y_np.shape[0]} historical rows from store")
        except FileNotFoundError:
            pass

    results = []
    scorer_cache: dict = {}
    drifted = True  # first day always trains
    for day in range(days):
        skip = retrain_policy == "drift" and not drifted
        r = run_cycle(
            state, store, n_rows, model_type=model_type,
            process_group=process_group, persist_fmt=persist_fmt,
            scorer_cache=scorer_cache, skip_train=skip,
        )
        results.append(r)
        offline_mape = (r["offline"] or {}).get("MAPE", float("inf"))
        drifted = r["online"]["MAPE"] > drift_threshold * offline_mape
        if process_group is not None:
            # collective decision: every rank must take the same
            # train/skip branch next cycle (a rank-local decision would
            # deadlock the training all-reduce); any drifted rank
            # triggers a fleet-wide retrain
            import torch.distributed as dist

            flag = torch.tensor(
                [1.0 if drifted else 0.0],
                device=device
                if dist.get_backend(process_group) == "nccl" else "cpu",
            )
            dist.all_reduce(flag, op=dist.ReduceOp.MAX, group=process_group)
            drifted = bool(flag.item() > 0)
        t = r["timings"]
        log.info(
            f"cycle {day + 1}/{days} ({state.date}): "
            f"{t['cycle_s']:.3f}s (train {t['train_s']:.3f} deploy "
            f"{t['deploy_s']:.3f} datagen {t['datagen_s']:.3f} "
            f"test {t['test_s']:.3f}) online MAPE "
            f"{r['online']['MAPE']:.4f}"
            + (" [retrain skipped: no drift]" if skip else "")
            + (" [drift detected]" if drifted and retrain_policy == "drift"
               else "")
        )
    state.drain_io()
    return results


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--store", default=None)
    p.add_argument("--days", type=int, default=30)
    p.add_argument("--rows", type=int, default=24 * 60)
    p.add_argument("--model", default="linear",
                   help="linear | poly[<degree>] | mlp | mlp-fp8 "
                        "(mlp with the MX-fp8 scoring forward)")
    p.add_argument("--device", default=None)
    p.add_argument("--start-date", default="2026-01-01")
    p.add_argument("--format", default="csv", choices=["csv", "npy"])
    p.add_argument("--json-out", default=None)
    p.add_argument("--retrain-policy", default="always",
                   choices=["always", "drift"])
    p.add_argument("--drift-threshold", type=float, default=1.5)
    args = p.parse_args(argv)
    if args.model == "mlp-fp8":  # env flag so the deployed scorer opts in
        import os

        os.environ["BODYWORK_MLP_FP8"] = "1"
        args.model = "mlp"
    results = run_loop(
        open_store(args.store), days=args.days, n_rows=args.rows,
        model_type=args.model, device=args.device,
        start_date=args.start_date, persist_fmt=args.format,
        retrain_policy=args.retrain_policy,
        drift_threshold=args.drift_threshold,
    )
    if args.json_out:
        with open(args.json_out, "w") as f:
            json.dump(results, f, indent=2, default=str)


if __name__ == "__main__":
    main()
