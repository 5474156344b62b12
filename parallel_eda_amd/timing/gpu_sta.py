"""GPU STA: levelized slack/criticality sweeps as HIP kernels.

Host side of csrc/hip/sta_kernel.hip. Graph topology (levels + CSRs) is
built once by the C++ TimingGraph and uploaded; per-iteration analysis
takes a device conn-delay tensor and produces device slack/crit tensors
(cpd returned to host).
"""
import ctypes as ct

import numpy as np

from ..arch.archdef import ArchDef
from ..ops import hip_api
from .. import ops


class StaLaunchArgs(ct.Structure):
    _fields_ = [
        ("level_blocks", ct.c_void_p), ("level_start", ct.c_void_p),
        ("in_ptr", ct.c_void_p), ("in_conn", ct.c_void_p),
        ("out_ptr", ct.c_void_p), ("out_conn", ct.c_void_p),
        ("conn_driver", ct.c_void_p), ("conn_sink", ct.c_void_p),
        ("is_seq", ct.c_void_p), ("blk_delay", ct.c_void_p),
        ("T_clb", ct.c_float), ("T_seq_out", ct.c_float),
        ("T_seq_in", ct.c_float), ("max_crit", ct.c_float),
        ("num_blocks", ct.c_int32), ("num_levels", ct.c_int32),
        ("num_conns", ct.c_int64),
        ("t_arr", ct.c_void_p), ("t_req", ct.c_void_p), ("cpd_out", ct.c_void_p),
        ("delay", ct.c_void_p), ("slack", ct.c_void_p), ("crit", ct.c_void_p),
        ("level_start_host", ct.c_void_p),
    ]


class StaDomainsArgs(ct.Structure):
    _fields_ = [
        ("base", StaLaunchArgs),
        ("block_clock", ct.c_void_p), ("periods", ct.c_void_p),
        ("K", ct.c_int32),
        ("arr_all", ct.c_void_p), ("req_all", ct.c_void_p),
        ("worst_bits", ct.c_void_p),
    ]


def _lib():
    lib = hip_api.lib()
    if not hasattr(lib, "_sta_ready"):
        lib.pnr_sta_analyze.restype = ct.c_int
        lib.pnr_sta_analyze.argtypes = [ct.POINTER(StaLaunchArgs), ct.c_void_p]
        lib.pnr_sta_args_sizeof.restype = ct.c_int64
        if lib.pnr_sta_args_sizeof() != ct.sizeof(StaLaunchArgs):
            raise RuntimeError("StaLaunchArgs ABI mismatch")
        lib.pnr_sta_analyze_domains.restype = ct.c_int
        lib.pnr_sta_analyze_domains.argtypes = [ct.POINTER(StaDomainsArgs),
                                                ct.c_void_p]
        lib.pnr_sta_domains_args_sizeof.restype = ct.c_int64
        if lib.pnr_sta_domains_args_sizeof() != ct.sizeof(StaDomainsArgs):
            raise RuntimeError("StaDomainsArgs ABI mismatch")
        lib._sta_ready = True
    return lib


class GpuSTA:
    def __init__(self, netlist, arch: ArchDef, device="cuda:0", max_crit=0.99):
        import torch
        self.torch = torch
        self.device = device
        self.arch = arch
        self.netlist = netlist
        self.max_crit = max_crit
        cpu = ops.cpu()
        from .sta import block_delays
        bd = block_delays(netlist, arch)
        self.tg = cpu.TimingGraph(
            netlist.cpp(), arch.T_clb, arch.T_seq_out, arch.T_seq_in,
            bd if bd is not None else np.empty(0, dtype=np.float32))
        self._blk_delay_host = bd
        blocks, start = self.tg.level_arrays()
        in_ptr, in_conn, out_ptr, out_conn, conn_driver = self.tg.csr_arrays()
        self.level_start_host = np.ascontiguousarray(start, dtype=np.int32)
        self.num_levels = self.tg.num_levels()

        def up(a):
            return torch.from_numpy(np.ascontiguousarray(a)).to(device)

        self.t_level_blocks = up(blocks)
        self.t_level_start = up(start)
        self.t_in_ptr = up(in_ptr); self.t_in_conn = up(in_conn)
        self.t_out_ptr = up(out_ptr); self.t_out_conn = up(out_conn)
        self.t_conn_driver = up(conn_driver)
        self.t_conn_sink = up(netlist.net_sinks)
        self.t_is_seq = up(netlist.block_is_seq)
        # heterogeneous: per-block comb delay on device; homogeneous: null
        # => the validated scalar-T_clb kernel path runs unchanged
        self.t_blk_delay = (up(self._blk_delay_host)
                            if self._blk_delay_host is not None else None)
        nb = netlist.num_blocks
        nc = netlist.num_conns
        self.t_arr = torch.zeros(nb, dtype=torch.float32, device=device)
        self.t_req = torch.zeros(nb, dtype=torch.float32, device=device)
        self.t_cpd = torch.zeros(1, dtype=torch.float32, device=device)
        self.t_slack = torch.zeros(nc, dtype=torch.float32, device=device)
        self.t_crit = torch.zeros(nc, dtype=torch.float32, device=device)
        self.lib = _lib()

    @property
    def num_conns(self):
        return self.netlist.num_conns

    def analyze_device(self, t_conn_delay):
        """Device path: conn delays in, (cpd, slack_t, crit_t) out
        (slack/crit stay on device)."""
        a = StaLaunchArgs()
        pt = lambda t: ct.c_void_p(t.data_ptr())
        a.level_blocks = pt(self.t_level_blocks)
        a.level_start = pt(self.t_level_start)
        a.in_ptr = pt(self.t_in_ptr); a.in_conn = pt(self.t_in_conn)
        a.out_ptr = pt(self.t_out_ptr); a.out_conn = pt(self.t_out_conn)
        a.conn_driver = pt(self.t_conn_driver)
        a.conn_sink = pt(self.t_conn_sink)
        a.is_seq = pt(self.t_is_seq)
        a.blk_delay = (pt(self.t_blk_delay)
                       if self.t_blk_delay is not None else None)
        a.T_clb = self.arch.T_clb; a.T_seq_out = self.arch.T_seq_out
        a.T_seq_in = self.arch.T_seq_in; a.max_crit = self.max_crit
        a.num_blocks = self.netlist.num_blocks
        a.num_levels = self.num_levels
        a.num_conns = self.netlist.num_conns
        a.t_arr = pt(self.t_arr); a.t_req = pt(self.t_req)
        a.cpd_out = pt(self.t_cpd)
        a.delay = pt(t_conn_delay); a.slack = pt(self.t_slack)
        a.crit = pt(self.t_crit)
        a.level_start_host = self.level_start_host.ctypes.data_as(ct.c_void_p)
        stream = self.torch.cuda.current_stream().cuda_stream
        rc = self.lib.pnr_sta_analyze(ct.byref(a), stream)
        hip_api.check(rc, "sta_analyze")
        return self.t_cpd, self.t_slack, self.t_crit

    def analyze_domains(self, conn_delay, block_clock, periods):
        """Multi-clock analysis on the device (reference:
        do_timing_analysis_new's (src,sink)-domain-pair loop;
        EXPERIMENTAL — pending round-2 GPU validation against the CPU
        TimingGraph.analyze_domains). Returns (worst_achieved_period,
        slack[], crit[]); the per-conn slack/crit arrays are exact, the
        worst-period scalar is tracked with a benign race and should be
        treated as advisory on the GPU path."""
        torch = self.torch
        K = len(periods)
        nb = self.netlist.num_blocks
        t_delay = torch.from_numpy(
            np.ascontiguousarray(conn_delay, dtype=np.float32)).to(self.device)
        t_bc = torch.from_numpy(
            np.ascontiguousarray(block_clock, dtype=np.int32)).to(self.device)
        t_per = torch.from_numpy(
            np.ascontiguousarray(periods, dtype=np.float32)).to(self.device)
        t_arr_all = torch.zeros(K * nb, dtype=torch.float32,
                                device=self.device)
        t_req_all = torch.zeros(K * nb, dtype=torch.float32,
                                device=self.device)
        t_worst = torch.zeros(2, dtype=torch.int32, device=self.device)
        d = StaDomainsArgs()
        d.base = self._base_args(t_delay)
        pt = lambda t: ct.c_void_p(t.data_ptr())
        d.block_clock = pt(t_bc); d.periods = pt(t_per); d.K = K
        d.arr_all = pt(t_arr_all); d.req_all = pt(t_req_all)
        d.worst_bits = pt(t_worst)
        stream = self.torch.cuda.current_stream().cuda_stream
        rc = self.lib.pnr_sta_analyze_domains(ct.byref(d), stream)
        hip_api.check(rc, "sta_analyze_domains")
        torch.cuda.synchronize(self.device)
        worst = float(np.frombuffer(
            t_worst[1:2].cpu().numpy().tobytes(), dtype=np.float32)[0])
        if worst == 0.0:
            worst = float(max(periods))
        return (worst, self.t_slack.cpu().numpy(),
                self.t_crit.cpu().numpy())

    def _base_args(self, t_conn_delay):
        a = StaLaunchArgs()
        pt = lambda t: ct.c_void_p(t.data_ptr())
        a.level_blocks = pt(self.t_level_blocks)
        a.level_start = pt(self.t_level_start)
        a.in_ptr = pt(self.t_in_ptr); a.in_conn = pt(self.t_in_conn)
        a.out_ptr = pt(self.t_out_ptr); a.out_conn = pt(self.t_out_conn)
        a.conn_driver = pt(self.t_conn_driver)
        a.conn_sink = pt(self.t_conn_sink)
        a.is_seq = pt(self.t_is_seq)
        a.blk_delay = (pt(self.t_blk_delay)
                       if self.t_blk_delay is not None else None)
        a.T_clb = self.arch.T_clb; a.T_seq_out = self.arch.T_seq_out
        a.T_seq_in = self.arch.T_seq_in; a.max_crit = self.max_crit
        a.num_blocks = self.netlist.num_blocks
        a.num_levels = self.num_levels
        a.num_conns = self.netlist.num_conns
        a.t_arr = pt(self.t_arr); a.t_req = pt(self.t_req)
        a.cpd_out = pt(self.t_cpd)
        a.delay = pt(t_conn_delay); a.slack = pt(self.t_slack)
        a.crit = pt(self.t_crit)
        a.level_start_host = self.level_start_host.ctypes.data_as(ct.c_void_p)
        return a

    def analyze(self, conn_delay):
        """Host-compatible API (numpy in/out), mirroring timing.sta.STA
        but with crit pre-clamped to max_crit."""
        torch = self.torch
        t_delay = torch.from_numpy(
            np.ascontiguousarray(conn_delay, dtype=np.float32)).to(self.device)
        self.analyze_device(t_delay)
        torch.cuda.synchronize(self.device)
        return (float(self.t_cpd.item()), self.t_slack.cpu().numpy(),
                self.t_crit.cpu().numpy())
