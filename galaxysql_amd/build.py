"""Build driver for the HIP extension (in-tree .so so it travels to the GPU
box with the repo snapshot) and the CPU oracle.

Usage: python -m galaxysql_amd.build
"""
import os
import subprocess
import sys

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HIPCC = os.environ.get("HIPCC", "hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def build_hip(verbose=True):
    src = os.path.join(CSRC, "gxhip.hip")
    out = os.path.join(CSRC, "libgxhip.so")
    incs = [os.path.join(src, "..", "..", "..")]
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-munsafe-fp-atomics",  # f64 atomicAdd -> global_atomic_add_f64
        "-fPIC", "-shared", src, "-o", out,
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True, cwd=CSRC)
    return out


def build_oracle(verbose=True):
    subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                   capture_output=not verbose)
    return os.path.join(REPO, "oracle", "libgxoracle.so")


def build_driver(verbose=True):
    subprocess.run(["make", "-C", os.path.join(REPO, "tools")], check=True,
                   capture_output=not verbose)
    return os.path.join(REPO, "tools", "gx_driver")


def build_all():
    build_oracle()
    build_hip()
    build_driver()


if __name__ == "__main__":
    build_all()
    print("built:", os.path.join(CSRC, "libgxhip.so"))
