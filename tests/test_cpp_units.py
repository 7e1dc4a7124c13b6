"""Compile-and-run C++ unit tests for the torch-free native pieces
(reference keeps a C++ test suite under test/cpp)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(240)
@pytest.mark.parametrize("sanitizer", [None, "address", "undefined"])
def test_cpp_shm_queue(tmp_path, sanitizer):
    src = os.path.join(ROOT, "tests", "cpp", "test_shm_queue.cpp")
    obj = os.path.join(ROOT, "glt_amd", "csrc", "cpu", "shm_queue.cpp")
    exe = str(tmp_path / f"test_shm_queue_{sanitizer}")
    flags = [f"-fsanitize={sanitizer}"] if sanitizer else ["-O2"]
    subprocess.run(
        ["g++", "-g", "-std=c++17", *flags, src, obj, "-o", exe,
         "-lpthread"], check=True)
    out = subprocess.run([exe], capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    assert "OK" in out.stdout
