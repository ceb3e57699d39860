"""Per-stage isolated environments — the reference's per-stage pip lists.

The reference installs a *different* pinned pip list in each stage's
container at startup (``bodywork.yaml:10-16,29-35,47-53,66-72`` —
deliberately divergent versions, e.g. numpy 1.19.5 in stages 1/3 vs
1.19.4 in stages 2/4).  The runner's fast path only VALIDATES declared
requirements against the shared environment
(``PipelineRunner.check_requirements``); this module supplies the full
execution mode: each distinct requirements list gets its own cached venv
(``--system-site-packages`` so the pinned framework stack stays visible)
with the stage's pins installed from an offline wheelhouse, and the
stage subprocess runs on that venv's interpreter.

Offline detail: ``python -m venv`` cannot bootstrap pip without network
(ensurepip is not shipped complete in this image), so venvs are created
``--without-pip`` and the pins are installed by the HOST interpreter's
pip with ``--no-index --find-links <wheelhouse> --target <venv
site-packages>`` — the venv's site-packages precedes the system's on
``sys.path``, so the stage resolves its own pinned versions first.
"""
from __future__ import annotations

import hashlib
import os
import subprocess
import sys
import sysconfig

from bodywork_mlops_demo_amd.utils.logging import configure_logger

log = configure_logger(__name__)

DEFAULT_WHEELHOUSE = "/opt/wheelhouse"


class StageEnvManager:
    """Builds and caches one venv per distinct requirements list."""

    def __init__(self, cache_dir: str | None = None,
                 wheelhouse: str | None = None):
        self.cache_dir = cache_dir or os.environ.get(
            "BODYWORK_AMD_ENV_CACHE",
            os.path.join(os.path.expanduser("~"), ".bodywork-amd-envs"))
        self.wheelhouse = wheelhouse or os.environ.get(
            "BODYWORK_AMD_WHEELHOUSE", DEFAULT_WHEELHOUSE)

    @staticmethod
    def _key(requirements: list[str]) -> str:
        canon = "\n".join(sorted(r.strip() for r in requirements if r.strip()))
        return hashlib.sha256(canon.encode()).hexdigest()[:16]

    def _site_packages(self, env_dir: str) -> str:
        ver = f"python{sys.version_info.major}.{sys.version_info.minor}"
        return os.path.join(env_dir, "lib", ver, "site-packages")

    def python_for(self, requirements: list[str]) -> str:
        """Interpreter path for a stage with these requirements; builds
        the venv on first use, reuses it afterwards.  No requirements ->
        the host interpreter."""
        reqs = [r.strip() for r in requirements if r.strip()]
        if not reqs:
            return sys.executable
        env_dir = os.path.join(self.cache_dir, self._key(reqs))
        py = os.path.join(env_dir, "bin", "python")
        stamp = os.path.join(env_dir, ".requirements.txt")
        want = "\n".join(sorted(reqs)) + "\n"
        if os.path.exists(py) and os.path.exists(stamp):
            with open(stamp) as f:
                if f.read() == want:
                    return py
        os.makedirs(self.cache_dir, exist_ok=True)
        log.info(f"building stage env {env_dir} for {len(reqs)} pins")
        subprocess.run(
            [sys.executable, "-m", "venv", "--without-pip",
             "--system-site-packages", "--clear", env_dir],
            check=True, capture_output=True, text=True)
        cmd = [sys.executable, "-m", "pip", "install", "--no-index",
               "--quiet", "--target", self._site_packages(env_dir)]
        if os.path.isdir(self.wheelhouse):
            cmd += ["--find-links", self.wheelhouse]
        proc = subprocess.run(cmd + reqs, capture_output=True, text=True)
        if proc.returncode != 0:
            raise RuntimeError(
                f"stage env install failed for {reqs}: "
                f"{proc.stderr[-2000:]}")
        with open(stamp, "w") as f:
            f.write(want)
        return py


def sys_site_packages() -> str:
    return sysconfig.get_paths()["purelib"]
