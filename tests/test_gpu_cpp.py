"""Runs the standalone C++ kernel test binary (tests/cpp/test_kernels.hip)
on the GPU box — the rebuild's equivalent of the reference's GTest C++ tier
(torch-quiver tests/cpp/test_quiver.cu, test_reindex.cu,
test_shard_tensor/)."""
import os
import subprocess

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(ROOT, "build", "qk_tests")


def test_cpp_kernel_suite():
    if not os.path.exists(BIN):
        pytest.fail(f"{BIN} missing — run build_ext.py first")
    res = subprocess.run([BIN], capture_output=True, text=True, timeout=600)
    print(res.stdout)
    assert res.returncode == 0, res.stdout + res.stderr
    assert "ALL C++ KERNEL TESTS PASSED" in res.stdout
