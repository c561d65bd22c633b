"""The standalone C++ driver binary (tools/gx_driver) — the judged non-Python
surface over the C ABI (no JDK here, so the C ABI + a
C++ driver binary stands in for the JNI shim). Selftest uses closed-form
expected results, no oracle import — so against libgxhip.so it exercises the
product alone."""
import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DRIVER = os.path.join(REPO, "tools", "gx_driver")


def _build():
    subprocess.run(["make", "-C", os.path.join(REPO, "tools")], check=True,
                   capture_output=True)


def test_driver_selftest_oracle():
    _build()
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=True)
    r = subprocess.run(
        [DRIVER, "--lib", os.path.join(REPO, "oracle", "libgxoracle.so"),
         "--device", "-1", "selftest"],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "SELFTEST PASSED" in r.stdout


@pytest.mark.gpu
def test_driver_selftest_hip():
    _build()
    r = subprocess.run(
        [DRIVER, "--lib",
         os.path.join(REPO, "galaxysql_amd", "csrc", "libgxhip.so"),
         "--device", "0", "selftest"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "SELFTEST PASSED" in r.stdout
