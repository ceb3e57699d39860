"""Per-stage environment isolation: two stages in one DAG resolve
DIFFERENT pinned versions of the same package (the reference installs a
divergent pip list per stage — bodywork.yaml numpy 1.19.5 in stages 1/3
vs 1.19.4 in stages 2/4)."""
import os
import zipfile

import pytest

from bodywork_mlops_demo_amd.config import load_config
from bodywork_mlops_demo_amd.pipeline.envs import StageEnvManager
from bodywork_mlops_demo_amd.pipeline.runner import PipelineRunner


def _make_wheel(wheelhouse: str, name: str, version: str) -> None:
    """Minimal valid pure-python wheel, built offline with zipfile."""
    di = f"{name}-{version}.dist-info"
    files = {
        f"{name}/__init__.py": f'__version__ = "{version}"\n',
        f"{di}/METADATA": (
            f"Metadata-Version: 2.1\nName: {name}\nVersion: {version}\n"),
        f"{di}/WHEEL": ("Wheel-Version: 1.0\nGenerator: test\n"
                        "Root-Is-Purelib: true\nTag: py3-none-any\n"),
    }
    record = "".join(f"{p},,\n" for p in files) + f"{di}/RECORD,,\n"
    files[f"{di}/RECORD"] = record
    path = os.path.join(wheelhouse, f"{name}-{version}-py3-none-any.whl")
    with zipfile.ZipFile(path, "w") as z:
        for p, content in files.items():
            z.writestr(p, content)


@pytest.fixture()
def wheelhouse(tmp_path):
    wh = tmp_path / "wheelhouse"
    wh.mkdir()
    _make_wheel(str(wh), "bodyworkdummy", "1.0.0")
    _make_wheel(str(wh), "bodyworkdummy", "2.0.0")
    return str(wh)


def test_env_manager_builds_and_caches(tmp_path, wheelhouse, monkeypatch):
    mgr = StageEnvManager(cache_dir=str(tmp_path / "envs"),
                          wheelhouse=wheelhouse)
    py1 = mgr.python_for(["bodyworkdummy==1.0.0"])
    py2 = mgr.python_for(["bodyworkdummy==2.0.0"])
    assert py1 != py2 and os.path.exists(py1) and os.path.exists(py2)
    # cached: second resolve returns the same interpreter without rebuild
    assert mgr.python_for(["bodyworkdummy==1.0.0"]) == py1
    # no requirements -> host interpreter
    import sys

    assert mgr.python_for([]) == sys.executable
    # the venv interpreter sees BOTH its pin and the system stack
    import subprocess

    out = subprocess.run(
        [py1, "-c", "import bodyworkdummy, numpy; "
                    "print(bodyworkdummy.__version__, numpy.__version__)"],
        capture_output=True, text=True, check=True).stdout
    assert out.startswith("1.0.0 ")


@pytest.mark.timeout(300)
def test_two_stages_resolve_different_pins(tmp_path, wheelhouse, monkeypatch):
    """The VERDICT acceptance test: a runner DAG where stage A pins
    bodyworkdummy==1.0.0 and stage B pins ==2.0.0, each stage reporting
    the version its interpreter actually imported."""
    monkeypatch.setenv("BODYWORK_AMD_ENV_CACHE", str(tmp_path / "envs"))
    monkeypatch.setenv("BODYWORK_AMD_WHEELHOUSE", wheelhouse)
    out_file = tmp_path / "versions.txt"
    cfg = load_config({
        "version": "1.0",
        "project": {"name": "env-iso", "DAG": "stage-a >> stage-b"},
        "stages": {
            "stage-a": {
                "executable_module_path": "tools/report_pkg.py",
                "args": ["--package", "bodyworkdummy", "--out",
                         str(out_file), "--tag", "a"],
                "requirements": ["bodyworkdummy==1.0.0"],
                "batch": {"max_completion_time_seconds": 120, "retries": 0},
            },
            "stage-b": {
                "executable_module_path": "tools/report_pkg.py",
                "args": ["--package", "bodyworkdummy", "--out",
                         str(out_file), "--tag", "b"],
                "requirements": ["bodyworkdummy==2.0.0"],
                "batch": {"max_completion_time_seconds": 120, "retries": 0},
            },
        },
    })
    runner = PipelineRunner(cfg, store_uri=str(tmp_path / "store"),
                            n_gpus=0, isolate_envs=True)
    report = runner.run()
    assert report.ok, report.failed
    lines = dict(ln.split("=") for ln in
                 out_file.read_text().strip().splitlines())
    assert lines == {"a": "1.0.0", "b": "2.0.0"}
