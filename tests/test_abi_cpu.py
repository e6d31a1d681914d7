"""C-ABI surface checks that run without a GPU: the library loads, exports
every symbol include/gpucompact.h declares, and fails LOUDLY (no CPU
fallback) when asked to compact with no HIP device."""
import ctypes
import os

import pytest

from conftest import REPO


def test_library_exports(product_lib):
    """Every function include/gpucompact.h declares must be exported."""
    import re
    lib = ctypes.CDLL(product_lib)
    hdr = open(os.path.join(REPO, "include", "gpucompact.h")).read()
    syms = set(re.findall(r"^\s*(?:int|const char\*|int32_t)\s+(gpuc_\w+)\s*\(",
                          hdr, re.M))
    assert {"gpuc_compact", "gpuc_generate", "gpuc_verify", "gpuc_flush",
            "gpuc_scrub", "gpuc_version", "gpuc_device_count"} <= syms, syms
    for sym in sorted(syms):
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
                  "token_lo", "n_overlaps", "gc_before", "cancel_flag",
                  "GPUC_ERR_CANCELLED"]:
        assert field in hdr
