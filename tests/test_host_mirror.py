"""C++ host-mirror binary (include/tnc_host.hpp): pinned-value assertions on
CPU; device contraction under -m gpu."""

import os
import subprocess

import pytest

BIN = os.path.join(os.path.dirname(__file__), "native", "host_mirror_test.bin")


@pytest.mark.skipif(not os.path.exists(BIN), reason="host mirror not built")
def test_host_mirror_cpu():
    out = subprocess.run([BIN], capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert "CPU assertions OK" in out.stdout


@pytest.mark.gpu
def test_host_mirror_gpu():
    assert os.path.exists(BIN), "host mirror binary not built"
    out = subprocess.run([BIN, "gpu"], capture_output=True, text=True,
                         timeout=300)
    assert out.returncode == 0, out.stderr
    assert "GPU contraction OK" in out.stdout
