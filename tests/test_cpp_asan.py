"""ASAN/UBSAN job for the native C++ components (SURVEY §5 sanitizer
requirement): builds tests/asan/core_asan_test.cpp (which includes
core.cpp) with -fsanitize=address,undefined and runs it; any leak,
overflow or UB fails the test. The capi header-only client compiles under
the same sanitizers as a second unit."""
import os
import subprocess
import sys
import sysconfig

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _build(src, out, extra=None):
    import pybind11
    inc = sysconfig.get_paths()["include"]
    cmd = ["g++", "-std=c++17", "-O1", "-g",
           "-fsanitize=address,undefined", "-fno-omit-frame-pointer",
           f"-I{inc}", f"-I{pybind11.get_include()}",
           src, "-o", out,
           f"-lpython{sys.version_info.major}.{sys.version_info.minor}",
           "-lpthread"] + (extra or [])
    r = subprocess.run(cmd, capture_output=True, text=True, cwd=ROOT)
    assert r.returncode == 0, r.stderr[-3000:]


@pytest.mark.timeout(300)
def test_core_under_asan(tmp_path):
    out = str(tmp_path / "core_asan_test")
    _build("tests/asan/core_asan_test.cpp", out)
    r = subprocess.run([out], capture_output=True, text=True,
                       env=dict(os.environ,
                                ASAN_OPTIONS="detect_leaks=1"))
    assert r.returncode == 0, r.stdout + r.stderr[-3000:]
    assert "OK" in r.stdout


@pytest.mark.timeout(300)
def test_capi_client_compiles_under_asan(tmp_path):
    """The header-only C++ client (capi/dynamo_client.hpp) must at least
    compile cleanly under the sanitizers (it is exercised live by
    test_cpp_client.py)."""
    out = str(tmp_path / "capi_demo")
    _build("capi/dynamo_client_demo.cpp", out)
    assert os.path.exists(out)
