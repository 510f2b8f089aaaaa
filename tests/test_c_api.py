"""C ABI test: compile tests/c_api/c_interface_test.c with gcc against
include/quda_amd.h, link libquda_amd_c.so, run it (role of the reference's
tests/c_interface_test). Runs the full Wilson-clover solve stack through
extern "C" symbols; on a GPU box the same binary exercises the HIP path
(QUDA_AMD_DEVICE unset -> cuda)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(ROOT, "libquda_amd_c.so")
SRC = os.path.join(ROOT, "tests", "c_api", "c_interface_test.c")


@pytest.fixture(scope="module")
def c_test_bin(tmp_path_factory):
    if not os.path.exists(SO):
        from build_hip import build_c_api
        sys.path.insert(0, ROOT)
        build_c_api()
    exe = str(tmp_path_factory.mktemp("c_api") / "c_interface_test")
    r = subprocess.run(
        ["gcc", "-std=c99", "-O2", SRC, "-o", exe, f"-L{ROOT}",
         "-lquda_amd_c", f"-Wl,-rpath,{ROOT}", "-lm"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    return exe


def _run(exe, device):
    env = dict(os.environ, QUDA_AMD_DEVICE=device)
    return subprocess.run([exe], env=env, capture_output=True, text=True,
                          timeout=600)


def test_c_interface_cpu(c_test_bin):
    r = _run(c_test_bin, "cpu")
    assert r.returncode == 0, (r.stdout, r.stderr)
    assert "ALL PASSED" in r.stdout


@pytest.mark.gpu
def test_c_interface_gpu(c_test_bin):
    r = _run(c_test_bin, "cuda:0")
    assert r.returncode == 0, (r.stdout, r.stderr)
    assert "ALL PASSED" in r.stdout


def test_fortran_interface_symbols(c_test_bin, tmp_path):
    """The trailing-underscore Fortran entry points (csrc/
    quda_fortran_api.cpp, role of lib/quda_fortran.F90) drive a full
    solve through pass-by-reference calls."""
    exe = str(tmp_path / "fortran_interface_test")
    src = os.path.join(ROOT, "tests", "c_api", "fortran_interface_test.c")
    r = subprocess.run(
        ["gcc", "-std=c99", "-O2", src, "-o", exe, f"-L{ROOT}",
         "-lquda_amd_c", f"-Wl,-rpath,{ROOT}", "-lm"],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = _run(exe, "cpu")
    assert out.returncode == 0, (out.returncode, out.stdout, out.stderr)
    assert "ALL PASSED" in out.stdout
