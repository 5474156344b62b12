"""Property-based tests (hypothesis) for pure helper invariants."""
import numpy as np
from hypothesis import given, settings, strategies as st


@settings(max_examples=60, deadline=None)
@given(st.integers(2, 40), st.integers(1, 6), st.integers(0, 10**6))
def test_bb_vectorization_matches_naive(n_nets, max_sinks, seed):
    """GpuRouter._compute_bbs' reduceat vectorization == the naive loop
    (the round-1 implementation) for arbitrary net/sink shapes,
    including empty sink lists."""
    rng = np.random.default_rng(seed)
    nx = ny = 20
    n_nodes = 500
    xlow = rng.integers(0, nx + 2, n_nodes).astype(np.int16)
    ylow = rng.integers(0, ny + 2, n_nodes).astype(np.int16)
    src = rng.integers(0, n_nodes, n_nets).astype(np.int32)
    counts = rng.integers(0, max_sinks + 1, n_nets)
    sink_ptr = np.r_[0, np.cumsum(counts)].astype(np.int32)
    sinks = rng.integers(0, n_nodes, sink_ptr[-1]).astype(np.int32)
    margin = rng.integers(0, 8, n_nets).astype(np.int32)

    class G:
        pass

    class A:
        pass

    g = G(); g.xlow = xlow; g.ylow = ylow
    arch = A(); arch.nx = nx; arch.ny = ny

    from parallel_eda_amd.route.gpu_router import GpuRouter
    r = object.__new__(GpuRouter)   # no __init__: pure helper under test
    r.g = g; r.arch = arch
    r.num_nets = n_nets
    r.src_rr = src; r.sink_ptr = sink_ptr; r.sink_rr = sinks
    r.bb_margin_per_net = margin
    fast = r._compute_bbs()

    ref = np.zeros((n_nets, 4), dtype=np.int16)
    for n in range(n_nets):
        terms = np.r_[src[n], sinks[sink_ptr[n]:sink_ptr[n + 1]]]
        xs = xlow[terms]; ys = ylow[terms]
        m = margin[n]
        ref[n] = (max(0, xs.min() - m), max(0, ys.min() - m),
                  min(nx + 1, xs.max() + m), min(ny + 1, ys.max() + m))
    assert np.array_equal(fast, ref)


@settings(max_examples=40, deadline=None)
@given(st.integers(0, 10**6), st.integers(2, 60), st.integers(1, 9))
def test_path_codec_roundtrip(seed, n_nodes, deg):
    """encode/decode a random walk over a random CSR graph."""
    from parallel_eda_amd.utils.path_codec import PathCodec
    rng = np.random.default_rng(seed)
    counts = rng.integers(1, deg + 1, n_nodes)
    row_ptr = np.r_[0, np.cumsum(counts)].astype(np.int64)
    edge_dst = rng.integers(0, n_nodes, row_ptr[-1]).astype(np.int32)
    codec = PathCodec(row_ptr, edge_dst)
    path = [int(rng.integers(0, n_nodes))]
    for _ in range(int(rng.integers(0, 40))):
        u = path[-1]
        lo, hi = row_ptr[u], row_ptr[u + 1]
        path.append(int(edge_dst[lo + int(rng.integers(0, hi - lo))]))
    packed = codec.encode(path)
    out = codec.decode(path[0], packed)
    assert list(out) == path


@settings(max_examples=40, deadline=None)
@given(st.integers(1, 300), st.integers(0, 6))
def test_smoothing_band_row_stochastic(n, radius):
    from parallel_eda_amd.place.delay_matrix import smoothing_band
    s = smoothing_band(n, radius)
    assert s.shape == (n, n)
    assert np.allclose(s.sum(axis=1), 1.0, atol=1e-5)
    assert (s >= 0).all()


@settings(max_examples=40, deadline=None)
@given(st.integers(0, 10**6), st.integers(1, 60), st.integers(8, 40))
def test_wave_schedule_invariants(seed, n_nets, grid):
    """schedule_bb_waves: every net appears exactly once; within a wave
    the coarse-cell footprints are disjoint."""
    from parallel_eda_amd.route.gpu_router import schedule_bb_waves
    rng = np.random.default_rng(seed)
    x0 = rng.integers(0, grid, n_nets)
    y0 = rng.integers(0, grid, n_nets)
    bb = np.stack([x0, y0,
                   np.minimum(grid + 1, x0 + rng.integers(0, grid, n_nets)),
                   np.minimum(grid + 1, y0 + rng.integers(0, grid, n_nets))],
                  axis=1).astype(np.int16)
    areas = ((bb[:, 2] - bb[:, 0] + 1).astype(np.int64) *
             (bb[:, 3] - bb[:, 1] + 1))
    ids = np.arange(n_nets)
    waves = schedule_bb_waves(bb, ids, areas, grid, grid, cell=8)
    seen = np.concatenate(waves) if waves else np.zeros(0, dtype=int)
    assert sorted(seen.tolist()) == ids.tolist()
    for w in waves:
        cells = set()
        for n in w:
            cs = {(cx, cy)
                  for cx in range(bb[n, 0] // 8, bb[n, 2] // 8 + 1)
                  for cy in range(bb[n, 1] // 8, bb[n, 3] // 8 + 1)}
            assert not (cells & cs), "overlapping nets in one wave"
            cells |= cs


@settings(max_examples=40, deadline=None)
@given(st.integers(0, 10**6), st.integers(1, 200), st.integers(1, 8))
def test_spatial_partition_invariants(seed, n_nets, world):
    """spatial_partition: every net assigned to a valid rank; weighted
    split is contiguous in bb-center order."""
    from parallel_eda_amd.parallel.dist import spatial_partition
    rng = np.random.default_rng(seed)
    x0 = rng.integers(0, 60, n_nets)
    bb = np.stack([x0, np.zeros(n_nets, dtype=np.int64),
                   x0 + rng.integers(0, 10, n_nets),
                   np.full(n_nets, 5)], axis=1).astype(np.int16)
    w = rng.uniform(0.1, 10.0, n_nets)
    r = spatial_partition(bb, world, weight=w)
    assert r.min() >= 0 and r.max() < world
    assert len(r) == n_nets
    # weighted balance: no rank exceeds 2x its fair share + max element
    share = w.sum() / world
    for q in range(world):
        assert w[r == q].sum() <= 2 * share + w.max() + 1e-9


@settings(max_examples=40, deadline=None)
@given(st.integers(0, 10**6), st.integers(1, 500))
def test_xcd_order_is_permutation(seed, n):
    from parallel_eda_amd.route.gpu_router import xcd_interleaved_order
    rng = np.random.default_rng(seed)
    bb = rng.integers(0, 100, (n, 4)).astype(np.int16)
    order = rng.permutation(n)
    out = xcd_interleaved_order(bb, order)
    assert sorted(out.tolist()) == sorted(order.tolist())
