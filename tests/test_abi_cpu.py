"""C-ABI surface checks that run without a GPU: the library loads, exports
every symbol include/gpucompact.h declares, and fails LOUDLY (no CPU
fallback) when asked to compact with no HIP device."""
import ctypes
import os

import pytest

from conftest import REPO


def test_library_exports(product_lib):
    lib = ctypes.CDLL(product_lib)
    for sym in ["gpuc_compact", "gpuc_generate", "gpuc_version", "gpuc_device_count"]:
        assert hasattr(lib, sym), f"missing export {sym}"
    lib.gpuc_version.restype = ctypes.c_char_p
    assert b"gfx950" in lib.gpuc_version()


def test_no_gpu_fails_loudly(product_lib):
    import cassandra_amd as ca
    if ca.device_count() > 0:
        pytest.skip("GPU present; the no-GPU refusal is exercised on CPU boxes")
    with pytest.raises(ca.GpuCompactError):
        ca.compact(["/nonexistent/oa-1-big"], "/tmp/oa-9-big")


def test_header_matches_wrapper():
    hdr = open(os.path.join(REPO, "include", "gpucompact.h")).read()
    for field in ["input_uncompressed_bytes", "merged_counts", "dominant_kernel",
                  "token_lo", "n_overlaps", "gc_before"]:
        assert field in hdr
