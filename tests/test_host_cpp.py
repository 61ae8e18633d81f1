"""C++ host mirror (vega_amd/host): builds, links the C ABI, and its
self-test passes on the GPU box. On CPU boxes it must fail LOUDLY."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HOST = os.path.join(ROOT, "vega_amd", "host")
CSRC = os.path.join(ROOT, "vega_amd", "csrc")
BIN = os.path.join(HOST, "vega_cli")


def build_cli():
    if not os.path.exists(os.path.join(CSRC, "libvega_gpu.so")):
        subprocess.check_call(
            ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
             "-shared", "vega_kernels.hip", "vega_api.hip", "-o", "libvega_gpu.so"],
            cwd=CSRC)
    subprocess.check_call(
        ["g++", "-O2", "-std=c++17", "-Wall", "vega_cli.cpp", "-o", "vega_cli",
         "-L" + CSRC, "-lvega_gpu", "-Wl,-rpath,$ORIGIN/../csrc",
         "-Wl,-rpath,/opt/rocm/lib"], cwd=HOST)


def test_host_builds_and_fails_loudly_without_gpu():
    build_cli()
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present; loud-failure leg is for CPU boxes")
    p = subprocess.run([BIN, "selftest"], capture_output=True, text=True, timeout=120)
    assert p.returncode != 0
    assert "FATAL" in p.stderr or "failed" in p.stderr.lower()


@pytest.mark.gpu
def test_host_selftest_gpu():
    build_cli()
    p = subprocess.run([BIN, "selftest"], capture_output=True, text=True, timeout=600)
    print(p.stdout, p.stderr)
    assert p.returncode == 0, p.stdout + p.stderr
    assert "all green" in p.stdout
