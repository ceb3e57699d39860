"""Package CLI: ``python -m bodywork_mlops_demo_amd <command>``.

Commands:
    run        — execute a bodywork.yaml pipeline (the bodywork-core role)
    loop       — multi-"day" concept-drift loop (retrain+redeploy cycles)
    analytics  — metric-history drift report (the analytics notebook role)
    train / serve / datagen / loadtest — run a single stage
"""
from __future__ import annotations

import sys

COMMANDS = {
    "run": "bodywork_mlops_demo_amd.pipeline.runner",
    "loop": "bodywork_mlops_demo_amd.pipeline.loop",
    "analytics": "bodywork_mlops_demo_amd.monitoring.analytics",
    "train": "bodywork_mlops_demo_amd.stages.train",
    "serve": "bodywork_mlops_demo_amd.stages.serve",
    "datagen": "bodywork_mlops_demo_amd.stages.datagen",
    "loadtest": "bodywork_mlops_demo_amd.stages.loadtest",
}


def main() -> None:
    if len(sys.argv) < 2 or sys.argv[1] in ("-h", "--help"):
        print(__doc__)
        sys.exit(0)
    cmd = sys.argv[1]
    if cmd not in COMMANDS:
        print(f"unknown command {cmd!r}; choose from {sorted(COMMANDS)}")
        sys.exit(2)
    import importlib

    mod = importlib.import_module(COMMANDS[cmd])
    mod.main(sys.argv[2:])


if __name__ == "__main__":
    main()
