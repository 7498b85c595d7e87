"""CPU checks of the product C-ABI library (cubefs_amd/libgfrs.so):
loads, exports every symbol include/gfrs.h declares, GPU-free entry points
match the oracle, and compute calls fail loudly without a GPU."""
import ctypes
import os
import re

import numpy as np
import pytest

from cubefs_amd import runtime
from cubefs_amd.runtime import Tactic

HDR = os.path.join(os.path.dirname(__file__), "..", "include", "gfrs.h")


def test_library_loads():
    L = runtime.lib()
    assert b"gfx950" in L.gfrs_version()


def test_all_header_symbols_exported():
    with open(HDR) as f:
        text = f.read()
    # function declarations: "type gfrs_xxx(" at top level
    decls = sorted(set(re.findall(r"\b(gfrs_[a-z0-9_]+)\s*\(", text)))
    decls = [d for d in decls if d not in ("gfrs_ctx",)]
    L = runtime.lib()
    missing = [d for d in decls if not hasattr(L, d)]
    assert not missing, missing


def test_buffer_sizes_match_oracle(oracle):
    L = runtime.lib()
    for (n, m, l, mss) in [(6, 3, 0, 2048), (12, 4, 0, 2048), (6, 10, 2, 0),
                           (16, 20, 2, 2048)]:
        for ds in (1, 100, 12288, 8 << 20):
            t = Tactic(n, m, l, max(1, l and 2 or 1), n + 1, 0, mss)
            ss = ctypes.c_int64()
            eds = ctypes.c_int64()
            es = ctypes.c_int64()
            rc = L.gfrs_buffer_sizes(ctypes.byref(t), ds, ctypes.byref(ss),
                                     ctypes.byref(eds), ctypes.byref(es))
            assert rc == 0
            oss, oeds, oes = oracle.buffer_sizes(n, m, l, mss, ds)
            assert (ss.value, eds.value, es.value) == (oss, oeds, oes)


def test_crc_size_math_matches_oracle(oracle):
    L = runtime.lib()
    for bl in (4096, 65536, 1 << 20):
        for size in (1, 100, 65532, 200000):
            assert L.gfrs_crc32b_encode_size(size, bl) == \
                oracle.crc32b_encode_size(size, bl)
            enc = oracle.crc32b_encode_size(size, bl)
            assert L.gfrs_crc32b_decode_size(enc, bl) == size
    assert L.gfrs_crc32b_encode_size(100, 1000) == -10


@pytest.mark.parametrize("k,m", [(4, 2), (6, 3), (12, 4), (16, 20), (24, 8),
                                 (15, 12), (10, 4), (8, 1), (7, 1)])
def test_product_encode_matrix_matches_oracle(oracle, k, m):
    """The product's host-side matrix math (gfrs_gf.cpp) must be
    bit-identical to the oracle's restatement of reedsolomon.go:220-244."""
    L = runtime.lib()
    out = np.zeros((k + m, k), dtype=np.uint8)
    rc = L.gfrs_compute_encode_matrix(
        k, k + m, out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
    assert rc == 0
    assert np.array_equal(out, oracle.build_matrix(k, k + m))


def test_no_gpu_fails_loudly():
    """Without a GPU the engine must refuse, never fall back to CPU."""
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("GPU present")
    except ImportError:
        pass
    L = runtime.lib()
    assert L.gfrs_device_count() == 0
    t = Tactic(6, 3, 0, 1, 8, 0, 2048)
    ctx = L.gfrs_create(ctypes.byref(t), -1)
    assert not ctx
    assert b"no HIP device" in L.gfrs_last_error()


def test_replicate_mode_accepted():
    """ec.NewEncoder accepts replicate tactics (codemode Replica3); Encode
    is a no-op and Verify vacuously true (reedsolomon.go:442,784).  These
    entry points are GPU-free, so they are testable here."""
    import torch
    if torch.cuda.is_available():
        pytest.skip("covered by GPU suite on a box")
    # without a GPU create still fails (no-fallback rule), but the tactic
    # must pass validation: check via the python registry
    from cubefs_amd import codemode
    t = codemode.get_tactic("Replica3")
    assert t.is_valid() and t.is_replicate()
