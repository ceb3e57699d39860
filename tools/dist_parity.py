"""Multi-rank DP parity check — run under torchrun, any world size.

Verifies the distributed training path end-to-end on the REAL backend
(RCCL when CUDA is available, else gloo): every rank runs the full
train->serve->drift->test cycle on its own data shard, gradients/stats
all-reduce over the process group, and the resulting model must be
BIT-IDENTICAL on every rank — the all-reduced quantities are the same
numbers everywhere and the optimiser/solve is deterministic.

World may exceed the visible GPU count (two ranks share one MI355X via
modulo device mapping) — that is the 1-GPU rehearsal for the driver's
8-GPU scale run: it exercises RCCL comm setup, the graph-captured
all-reduce, and the shared artefact store under multi-process contention.

    torchrun --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 \
        tools/dist_parity.py --model linear --cycles 2

Prints one line per model: ``DIST_PARITY OK model=... world=N
max_diff=0.0``; exits non-zero on any cross-rank divergence.
"""
from __future__ import annotations

import argparse
import os
import shutil
import sys
import tempfile
from datetime import date

import torch
import torch.distributed as dist

# torchrun sets sys.path[0] to tools/; the package lives at the repo root
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _param_tensors(model) -> list[torch.Tensor]:
    if hasattr(model, "w1"):  # MLP
        return [model.w1, model.b1, model.W2, model.b2, model.w3, model.b3]
    if hasattr(model, "coef_t_"):  # poly (normalised-basis coefficients)
        return [torch.tensor([float(c) for c in model.coef_t_],
                             dtype=torch.float64)]
    return [torch.tensor([float(model.coef_), float(model.intercept_)])]


def check_model(model_type: str, args, device: str, rank: int, world: int,
                pg) -> float:
    from bodywork_mlops_demo_amd.pipeline.cycle import CycleState, run_cycle
    from bodywork_mlops_demo_amd.store import LocalStore

    store_dir = os.path.join(
        tempfile.gettempdir(),
        f"dist_parity_{os.environ.get('MASTER_PORT', '0')}_{model_type}")
    if rank == 0:
        shutil.rmtree(store_dir, ignore_errors=True)
        os.makedirs(store_dir, exist_ok=True)
    dist.barrier(group=pg)
    store = LocalStore(store_dir)

    state = CycleState(device, date(2026, 2, 1), rank=rank, world_size=world,
                       history_days=1)
    cache: dict = {}
    last = None
    for _ in range(args.cycles):
        last = run_cycle(
            state, store, args.rows, model_type=model_type,
            process_group=pg, persist_fmt="npy",
            mlp_steps=args.mlp_steps, mlp_batch_size=args.mlp_batch,
            use_graphs=device.startswith("cuda"), scorer_cache=cache,
        )
    state.drain_io()

    # Train once more OUTSIDE the cycle and compare the TRAINED model
    # across ranks (the deployed scorer re-reads rank 0's artefact, so it
    # is identical by construction and proves only the store round-trip;
    # the trained model proves the all-reduce itself: each rank holds a
    # different data shard, so identical weights require the collective).
    from bodywork_mlops_demo_amd.stages import train as stage1

    _, model = stage1.run(
        store, model_type=model_type, device=device, process_group=pg,
        rank=rank, world_size=1, mlp_steps=args.mlp_steps,
        mlp_batch_size=args.mlp_batch,
        data=(state.y, state.X, state.date), return_model=True,
        model_cache=cache,
    )
    max_diff = 0.0
    for t in _param_tensors(model):
        t = t.detach().float()
        t_dev = t.to(device) if not t.is_cuda and device.startswith("cuda") \
            else t
        ref = t_dev.clone()
        dist.broadcast(ref, src=0, group=pg)
        d = (t_dev - ref).abs().max().item()
        max_diff = max(max_diff, d)
    buf = torch.tensor([max_diff], dtype=torch.float64,
                       device=device if device.startswith("cuda") else "cpu")
    dist.all_reduce(buf, op=dist.ReduceOp.MAX, group=pg)
    max_diff = float(buf.item())

    if rank == 0:
        online = last["online"] if last else {}
        status = "OK" if max_diff == 0.0 else "DIVERGED"
        print(f"DIST_PARITY {status} model={model_type} world={world} "
              f"max_diff={max_diff} cycles={args.cycles} "
              f"online_mape={online.get('MAPE')}", flush=True)
    return max_diff


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="linear",
                   help="comma-separated: linear,poly3,mlp")
    p.add_argument("--cycles", type=int, default=2)
    p.add_argument("--rows", type=int, default=200_000)
    p.add_argument("--mlp-steps", type=int, default=10)
    p.add_argument("--mlp-batch", type=int, default=16384)
    p.add_argument("--backend", default=None,
                   help="force nccl|gloo (default: auto — nccl when every "
                        "rank can own a distinct GPU, else gloo transport "
                        "with GPU-resident compute; RCCL refuses 2 ranks "
                        "on 1 device)")
    args = p.parse_args()

    from bodywork_mlops_demo_amd.parallel import init_distributed

    rank, world, local_rank = init_distributed(backend=args.backend)
    if world < 2:
        raise SystemExit("run under torchrun with --nproc-per-node >= 2")
    use_cuda = torch.cuda.is_available()
    dev_idx = local_rank % torch.cuda.device_count() if use_cuda else 0
    device = f"cuda:{dev_idx}" if use_cuda else "cpu"
    if use_cuda:
        from bodywork_mlops_demo_amd import ops

        if not ops.hip_available():
            raise RuntimeError("HIP extension missing on a GPU box")
        torch.cuda.set_device(dev_idx)

    pg = dist.group.WORLD
    worst = 0.0
    for m in args.model.split(","):
        worst = max(worst, check_model(m.strip(), args, device, rank,
                                       world, pg))
    dist.barrier(group=pg)
    dist.destroy_process_group()
    sys.exit(0 if worst == 0.0 else 1)


if __name__ == "__main__":
    main()
