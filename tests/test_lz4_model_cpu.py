"""Fuzz the product's LZ4 model against the system liblz4 1.9.3 (the version
the reference bundles via lz4-java 1.8.0 — BASELINE.md parity anchors)."""
import os
import subprocess

from conftest import REPO


def test_lz4_model_bit_exact(tmp_path):
    binp = str(tmp_path / "lz4fuzz")
    subprocess.run(
        ["g++", "-O2", "-std=c++17", "-o", binp,
         os.path.join(REPO, "tests", "native", "lz4_model_fuzz.cpp"),
         "-l:liblz4.so.1"],
        check=True)
    r = subprocess.run([binp], capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "OK" in r.stdout
