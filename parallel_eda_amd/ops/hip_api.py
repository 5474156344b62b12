"""ctypes API over libpnr_hip.so (gfx950 kernels).

Struct layouts must match csrc/hip/router_kernel.hip exactly.
"""
import ctypes as ct

from . import hip


class RouteLaunchArgs(ct.Structure):
    _fields_ = [
        # RRDev
        ("type", ct.c_void_p), ("xlow", ct.c_void_p), ("ylow", ct.c_void_p),
        ("xhigh", ct.c_void_p), ("yhigh", ct.c_void_p), ("capacity", ct.c_void_p),
        ("R", ct.c_void_p), ("C", ct.c_void_p),
        ("row_ptr", ct.c_void_p), ("edge_dst", ct.c_void_p), ("edge_sw", ct.c_void_p),
        ("sw_R", ct.c_void_p), ("sw_Tdel", ct.c_void_p), ("base_cost", ct.c_void_p),
        ("idx_in_tile", ct.c_void_p),
        ("num_nodes", ct.c_int32), ("nx", ct.c_int32), ("ny", ct.c_int32),
        ("L", ct.c_int32), ("npt", ct.c_int32),
        # NetsDev
        ("net_src", ct.c_void_p), ("sink_ptr", ct.c_void_p), ("sink_rr", ct.c_void_p),
        ("crit", ct.c_void_p), ("sink_orig", ct.c_void_p), ("bb", ct.c_void_p),
        ("num_nets", ct.c_int32),
        # TreesDev
        ("tree_off", ct.c_void_p), ("tree_node", ct.c_void_p),
        ("tree_parent", ct.c_void_p), ("tree_sw", ct.c_void_p),
        ("tree_delay", ct.c_void_p), ("tree_len", ct.c_void_p),
        ("sink_delay", ct.c_void_p),
        # params
        ("astar_fac", ct.c_float), ("pres_fac", ct.c_float),
        ("seg_delay", ct.c_float), ("ipin_delay", ct.c_float),
        ("seg_base", ct.c_float), ("ipin_base", ct.c_float),
        ("delta_fac", ct.c_float),
        ("cong_mult", ct.c_float),
        ("max_rounds", ct.c_int32),
        ("strict_term", ct.c_int32),
        # queues
        ("queue_small", ct.c_void_p), ("n_queue_small", ct.c_int32),
        ("queue_large", ct.c_void_p), ("n_queue_large", ct.c_int32),
        ("q_cursors", ct.c_void_p),
        ("occ", ct.c_void_p), ("acc", ct.c_void_p),
        ("state_base", ct.c_void_p), ("small_cap", ct.c_int64),
        ("large_cap", ct.c_int64), ("n_small_slots", ct.c_int32),
        ("n_large_slots", ct.c_int32),
        ("frontier_base", ct.c_void_p), ("f_cap_small", ct.c_int64),
        ("f_cap_large", ct.c_int64),
        ("touched_base", ct.c_void_p), ("t_cap_small", ct.c_int64),
        ("t_cap_large", ct.c_int64),
        ("fail_flags", ct.c_void_p),
        ("stats", ct.c_void_p),
        ("net_scans", ct.c_void_p),
        ("use_calendar", ct.c_int32),
        ("partial", ct.c_int32),
        ("inq_base", ct.c_void_p),
    ]


_lib = None


def lib():
    global _lib
    if _lib is None:
        _lib = hip()
        _lib.pnr_route_nets.restype = ct.c_int
        _lib.pnr_route_nets.argtypes = [ct.POINTER(RouteLaunchArgs), ct.c_void_p]
        _lib.pnr_update_acc.restype = ct.c_int
        _lib.pnr_update_acc.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_void_p,
                                        ct.c_float, ct.c_int32, ct.c_void_p]
        _lib.pnr_overuse_count.restype = ct.c_int
        _lib.pnr_overuse_count.argtypes = [ct.c_void_p] * 4 + [ct.c_int32, ct.c_void_p]
        _lib.pnr_recount_occ.restype = ct.c_int
        _lib.pnr_recount_occ.argtypes = [ct.c_void_p] * 4 + [ct.c_int32, ct.c_void_p, ct.c_void_p]
        _lib.pnr_rip_up_nets.restype = ct.c_int
        _lib.pnr_rip_up_nets.argtypes = [ct.c_void_p] * 4 + [ct.c_int32, ct.c_void_p, ct.c_void_p]
        _lib.pnr_flag_congested_nets.restype = ct.c_int
        _lib.pnr_flag_congested_nets.argtypes = [ct.c_void_p] * 5 + [ct.c_int32, ct.c_void_p, ct.c_void_p]
        _lib.pnr_fill_u64_launch.restype = ct.c_int
        _lib.pnr_fill_u64_launch.argtypes = [ct.c_void_p, ct.c_uint64, ct.c_int64, ct.c_void_p]
        _lib.pnr_mwg_route_net.restype = ct.c_int
        _lib.pnr_mwg_route_net.argtypes = [
            ct.POINTER(RouteLaunchArgs), ct.c_int32, ct.c_int32, ct.c_int32,
            ct.c_void_p, ct.c_void_p, ct.c_void_p, ct.c_void_p, ct.c_int64,
            ct.c_void_p, ct.c_int32, ct.c_void_p]
        _lib.pnr_mfma_gemm_f32.restype = ct.c_int
        _lib.pnr_mfma_gemm_f32.argtypes = [ct.c_void_p, ct.c_void_p,
                                           ct.c_void_p, ct.c_int32,
                                           ct.c_int32, ct.c_int32,
                                           ct.c_void_p]
        _lib.pnr_route_args_sizeof.restype = ct.c_int64
        # ABI guard: the ctypes mirror must match the C struct exactly —
        # a silent mismatch turns into near-null GPU pointer faults.
        c_size = _lib.pnr_route_args_sizeof()
        py_size = ct.sizeof(RouteLaunchArgs)
        if c_size != py_size:
            raise RuntimeError(
                f"RouteLaunchArgs ABI mismatch: C {c_size} vs ctypes {py_size}")
    return _lib


def check(rc, what):
    if rc != 0:
        raise RuntimeError(f"HIP error {rc} in {what}")
