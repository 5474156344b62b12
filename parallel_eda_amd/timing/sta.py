"""Static timing analysis wrapper.

Reference semantics: vpr/SRC/timing/path_delay.c:1994 do_timing_analysis_new
(T_arr forward max-plus per level, T_req backward, slack, criticality) and
router glue router.cxx:27-40 analyze_timing. CPU engine in
csrc/cpu/sta_serial.cpp; GPU engine runs the same levelized sweeps as HIP
kernels (csrc/hip/sta_kernel.hip).
"""
import numpy as np

from ..arch.archdef import ArchDef
from .. import ops


def block_delays(netlist, arch: ArchDef):
    """Per-block combinational delay by block type (CLB/RAM/DSP), or None
    for a homogeneous arch (scalar T_clb path)."""
    if not arch.is_heterogeneous():
        return None
    bt = np.asarray(netlist.block_type)
    bd = np.full(len(bt), arch.T_clb, dtype=np.float32)
    for t in (2, 3):  # BLK_RAM, BLK_DSP
        bd[bt == t] = arch.block_delay_of(t)
    return bd


class STA:
    def __init__(self, netlist, arch: ArchDef):
        cpu = ops.cpu()
        self.netlist = netlist
        self.arch = arch
        bd = block_delays(netlist, arch)
        if bd is None:
            bd = np.empty(0, dtype=np.float32)
        self.tg = cpu.TimingGraph(netlist.cpp(), arch.T_clb, arch.T_seq_out,
                                  arch.T_seq_in, bd)

    @property
    def num_levels(self):
        return self.tg.num_levels()

    def analyze(self, conn_delay):
        """Returns (critical_path_delay, slack[], crit[]) per connection."""
        conn_delay = np.ascontiguousarray(conn_delay, dtype=np.float32)
        cpd, slack, crit = self.tg.analyze(conn_delay)
        return float(cpd), np.asarray(slack), np.asarray(crit)

    def analyze_domains(self, conn_delay, block_clock, periods,
                        pair_skip=None, pair_mult=None):
        """Multi-clock analysis (reference: do_timing_analysis_new per
        (src,sink)-domain pairs). block_clock: per-block domain id (-1
        comb); periods: seconds per domain. pair_skip / pair_mult:
        optional KxK (src,sink)-pair constraint arrays from SDC
        set_false_path / set_multicycle_path (reference read_sdc.c).
        Returns (worst_achieved_period, slack[], crit[]) — slack/crit
        are the worst/max over analyzed domain pairs."""
        conn_delay = np.ascontiguousarray(conn_delay, dtype=np.float32)
        bc = np.ascontiguousarray(block_clock, dtype=np.int32)
        pr = np.ascontiguousarray(periods, dtype=np.float32)
        if pair_skip is not None:
            pair_skip = np.ascontiguousarray(pair_skip, dtype=np.uint8)
        if pair_mult is not None:
            pair_mult = np.ascontiguousarray(pair_mult, dtype=np.float32)
        wp, slack, crit = self.tg.analyze_domains(conn_delay, bc, pr,
                                                  pair_skip, pair_mult)
        return float(wp), np.asarray(slack), np.asarray(crit)
