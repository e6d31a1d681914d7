"""CPU pins for the Snappy (C3) codec chain: the scalar 1.1.8 restatement vs
the SYSTEM libsnappy (the parity anchor — BASELINE.md caveat: the reference
bundles 1.1.10, format-stable), and the 64-probe wave-window decomposition
(the HIP kernel's control flow) vs that restatement. Both byte-for-byte."""
import os
import subprocess

import pytest

from conftest import REPO


def _run(binp, srcs, libs=()):
    subprocess.run(["g++", "-O2", "-std=c++17", "-o", binp, *srcs, *libs],
                   check=True)
    r = subprocess.run([binp], capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "OK" in r.stdout


def test_snappy_model_vs_libsnappy(tmp_path):
    if not (os.path.exists("/opt/conda/lib/libsnappy.so.1")
            or os.path.exists("/usr/lib/x86_64-linux-gnu/libsnappy.so.1")):
        pytest.skip("libsnappy not present")
    _run(str(tmp_path / "snp_model"),
         [os.path.join(REPO, "tests", "native", "snappy_model_test.cpp")],
         ["-ldl"])


def test_snappy_wave_decomposition_vs_model(tmp_path):
    _run(str(tmp_path / "snp_sim"),
         [os.path.join(REPO, "tests", "native", "snappy_sim_test.cpp")])
