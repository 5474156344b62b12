"""Router-measured placement delay matrix.

Reference: place/timing_place_lookup.c:981 compute_delay_lookup_tables —
the placer's (dx,dy) -> delay table is built by INVOKING THE ACTUAL ROUTER
on dummy nets. We route one single-sink net from a corner source tile to
every reachable tile with criticality 1.0 (pure-delay cost, congestion
term zeroed), so overlapping routes don't interact and the whole batch can
route concurrently on the GPU.
"""
import numpy as np

from ..arch.archdef import ArchDef
from .. import ops, rrgraph


def smoothing_band(n: int, radius: int) -> np.ndarray:
    """Row-normalized band-averaging matrix (distance-weighted)."""
    idx = np.arange(n)
    w = np.maximum(0.0, radius + 1 - np.abs(idx[:, None] - idx[None, :]))
    return (w / w.sum(axis=1, keepdims=True)).astype(np.float32)


def mfma_gemm(a, b):
    """C = A @ B on the MFMA pipes (f32 16x16x4 tiles;
    csrc/hip/mfma_kernels.hip). a, b: torch CUDA f32 2-D tensors."""
    import torch
    from ..ops import hip_api
    from ..route.gpu_router import ct_ptr
    assert a.dtype == torch.float32 and b.dtype == torch.float32
    a = a.contiguous(); b = b.contiguous()
    M, K = a.shape
    K2, N = b.shape
    assert K == K2
    c = torch.empty((M, N), dtype=torch.float32, device=a.device)
    lib = hip_api.lib()
    rc = lib.pnr_mfma_gemm_f32(
        ct_ptr(a), ct_ptr(b), ct_ptr(c), M, N, K,
        torch.cuda.current_stream(a.device).cuda_stream)
    hip_api.check(rc, "mfma_gemm_f32")
    return c


def smooth_delay_matrix_gpu(dm: np.ndarray, radius: int = 1,
                            device="cuda:0") -> np.ndarray:
    """R = S_r @ D @ S_c^T band smoothing of the routed delay matrix on
    the MFMA matrix cores (reference post-pass on the
    timing_place_lookup.c tables; the raw router-measured matrix has
    single-sample noise along the edge rows)."""
    import torch
    nr, nc = dm.shape
    sr = torch.from_numpy(smoothing_band(nr, radius)).to(device)
    sct = torch.from_numpy(smoothing_band(nc, radius).T.copy()).to(device)
    d = torch.from_numpy(np.ascontiguousarray(dm, dtype=np.float32)).to(device)
    t = mfma_gemm(sr, d)
    r = mfma_gemm(t, sct)
    return r.cpu().numpy()


def routed_delay_matrix(arch: ArchDef, g=None, engine="cpu", device="cuda:0",
                        smooth_radius=0):
    """Returns delay_mat shaped (nx+2, ny+2), indexed [|dx|, |dy|].

    smooth_radius > 0 (GPU engine): MFMA band smoothing of the measured
    table (smooth_delay_matrix_gpu)."""
    if g is None:
        g = rrgraph.build_rr_graph(arch)
    nx, ny = arch.nx, arch.ny
    gy = ny + 2
    ts = np.asarray(g.tile_source)
    tk = np.asarray(g.tile_sink)
    sx, sy = 1, 1
    src = ts[sx * gy + sy]
    targets = [(x, y) for x in range(1, nx + 1) for y in range(1, ny + 1)
               if (x, y) != (sx, sy)]
    n = len(targets)
    src_rr = np.full(n, src, dtype=np.int32)
    sink_ptr = np.arange(n + 1, dtype=np.int64)
    sink_rr = np.asarray([tk[x * gy + y] for x, y in targets], dtype=np.int32)
    crit = np.full(n, 1.0, dtype=np.float32)  # pure-delay routing

    if engine == "gpu":
        from ..route.gpu_router import GpuRouter
        r = GpuRouter(g, arch, src_rr, sink_ptr.astype(np.int32), sink_rr,
                      device=device)
        _, sd = r.route_iteration(crit, 0.0)
        delays = sd
    else:
        cpu = ops.cpu()
        opts = cpu.RouterOpts()
        r = cpu.SerialRouter(g, src_rr, sink_ptr, sink_rr, opts)
        r.set_pres_fac(0.0)
        r.route_iteration(crit)
        delays = np.asarray(r.sink_delays())

    dm = np.zeros((nx + 2, ny + 2), dtype=np.float32)
    cnt = np.zeros((nx + 2, ny + 2), dtype=np.int32)
    for (x, y), d in zip(targets, delays):
        dx, dy = abs(x - sx), abs(y - sy)
        dm[dx, dy] += d
        cnt[dx, dy] += 1
    with np.errstate(invalid="ignore"):
        dm = np.where(cnt > 0, dm / np.maximum(cnt, 1), 0.0)
    # same-tile connection: OPIN buffer + IPIN mux only
    dm[0, 0] = arch.T_opin + arch.T_ipin
    # extrapolate the unreached outer rows/cols (IO ring offsets) from the
    # last measured diagonal step
    for dx in range(nx + 2):
        for dy in range(ny + 2):
            if dx + dy > 0 and dm[dx, dy] == 0.0:
                base = dm[min(dx, nx - 1), min(dy, ny - 1)]
                extra = (max(0, dx - (nx - 1)) + max(0, dy - (ny - 1)))
                step = dm[1, 0] - dm[0, 0] if dm[1, 0] > 0 else 0.0
                dm[dx, dy] = base + extra * max(step, 0.0)
    dm = dm.astype(np.float32)
    if smooth_radius > 0 and engine == "gpu":
        dm = smooth_delay_matrix_gpu(dm, radius=smooth_radius,
                                     device=device)
    return dm
