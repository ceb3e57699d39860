"""The four pipeline stage workloads (reference L2).

Each stage is a library function (``run(...)``) plus an executable module
entry, mirroring the reference's one-module-per-stage layout
(``mlops_simulation/stage_[1-4]*.py``) while sharing the store / logging /
monitoring layers instead of copy-pasting them.
"""
