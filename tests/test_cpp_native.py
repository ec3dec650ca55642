"""Native C++ end-to-end chain over the C-ABI + GpuReplicator header
(include/rocksplicator_replicator.hpp): built and run as a subprocess."""
import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "build", "test_cpp_chain")


def build_bin():
    os.makedirs(os.path.join(REPO, "build"), exist_ok=True)
    subprocess.run(
        ["hipcc", "-O3", "-std=c++17", "scripts/test_cpp_chain.cpp",
         "-Iinclude", "-Lrocksplicator_amd", "-lgra",
         "-Wl,-rpath,$ORIGIN/../rocksplicator_amd", "-o", BIN],
        cwd=REPO, check=True)


def test_cpp_chain_builds():
    """Compile-check on CPU boxes (no GPU needed to build)."""
    build_bin()
    assert os.path.exists(BIN)


@pytest.mark.gpu
def test_cpp_chain_runs():
    if not os.path.exists(BIN):
        build_bin()
    r = subprocess.run([BIN], capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "cpp chain OK" in r.stdout
