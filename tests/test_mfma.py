"""MFMA matrix-core kernel numerics (gfx950 v_mfma_f32_16x16x4_f32).

Reference use: the routed placement delay matrix's band smoothing
(place/timing_place_lookup.c:981 tables; delay_matrix.py)."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_mfma_gemm_matches_numpy():
    import torch
    from parallel_eda_amd.place.delay_matrix import mfma_gemm
    rng = np.random.default_rng(7)
    # asymmetric shapes + values (a symmetric B would mask a row/col swap)
    for (m, n, k) in [(16, 16, 4), (16, 16, 16), (122, 202, 122),
                      (37, 65, 51), (282, 282, 282)]:
        a = rng.standard_normal((m, k)).astype(np.float32)
        b = rng.standard_normal((k, n)).astype(np.float32)
        c = mfma_gemm(torch.from_numpy(a).cuda(),
                      torch.from_numpy(b).cuda()).cpu().numpy()
        ref = a.astype(np.float64) @ b.astype(np.float64)
        # MFMA f32 is an exact fmaf chain; vs the float64 reference the
        # error grows with the K-long accumulation — scale the bound
        tol = 1e-6 + 4e-7 * k
        assert np.allclose(c, ref, rtol=tol, atol=tol), (m, n, k)


def test_mfma_delay_matrix_smoothing():
    import torch
    from parallel_eda_amd.place.delay_matrix import (
        smooth_delay_matrix_gpu, smoothing_band)
    rng = np.random.default_rng(3)
    dm = np.abs(rng.standard_normal((122, 122))).astype(np.float32)
    out = smooth_delay_matrix_gpu(dm, radius=2)
    sr = smoothing_band(122, 2).astype(np.float64)
    ref = sr @ dm.astype(np.float64) @ sr.T
    assert np.allclose(out, ref, rtol=1e-4, atol=1e-6)


def test_routed_delay_matrix_gpu_smoothed():
    """End-to-end: router-measured delay table on GPU + MFMA smoothing
    stays monotone-ish and close to the raw table."""
    from parallel_eda_amd.arch.archdef import get_arch
    from parallel_eda_amd import rrgraph
    from parallel_eda_amd.place.delay_matrix import routed_delay_matrix
    arch = get_arch("tseng")
    g = rrgraph.build_rr_graph(arch)
    raw = routed_delay_matrix(arch, g=g, engine="gpu")
    sm = routed_delay_matrix(arch, g=g, engine="gpu", smooth_radius=1)
    assert sm.shape == raw.shape
    # smoothing is an averaging: stays within the raw table's range and
    # close to it in the interior
    assert sm.min() >= 0
    assert sm.max() <= raw.max() * 1.0001
    mid = (slice(1, -2), slice(1, -2))
    assert np.abs(sm[mid] - raw[mid]).max() <= 0.5 * raw[mid].max()
