"""Host-side sanitizer pass (SURVEY §5 race-detection/sanitizers row):
the shared device headers' logic is compiled for the HOST with ASAN +
UBSAN and cross-checked word-for-word against the numpy philox oracle.
Any integer overflow, OOB access or UB in the header trips the build's
sanitizers and fails the run."""
import os
import subprocess

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLANGXX = "/opt/rocm/lib/llvm/bin/clang++"


@pytest.mark.skipif(not os.path.exists(CLANGXX),
                    reason="ROCm clang++ not present")
@pytest.mark.timeout(300)
def test_philox_header_under_asan_ubsan(tmp_path):
    exe = str(tmp_path / "philox_host")
    build = subprocess.run(
        [CLANGXX, "-O1", "-std=c++17", "-g", "-I/opt/rocm/include",
         "-fsanitize=address,undefined", "-fno-sanitize-recover=all",
         os.path.join(REPO, "tests", "asan", "philox_host.cpp"),
         "-o", exe],
        capture_output=True, text=True)
    assert build.returncode == 0, build.stderr[-2000:]
    run = subprocess.run([exe, "42", str(0x1F123BB5)],
                         capture_output=True, text=True)
    assert run.returncode == 0, (run.stdout + run.stderr)[-2000:]

    from bodywork_mlops_demo_amd.ops import reference

    for line in run.stdout.strip().splitlines():
        c, x, y, z, w = (int(v) for v in line.split())
        want = reference.philox4x32(
            np.array([c], dtype=np.uint64), 42, 0x1F123BB5)
        got = np.array([[x, y, z, w]], dtype=np.uint32)
        assert (want.astype(np.uint32) == got).all(), (c, want, got)
