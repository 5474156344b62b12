"""Post-route power estimation.

Reference scope: power/power.c:1695 power_total — dynamic + leakage per
component (routing, clock, logic blocks) from per-net switching
activities. This is the block-level subset matching our device model:

  P_dyn(net)  = 0.5 * a(net) * f_clk * Vdd^2 * C(net)
  C(net)      = sum of wire C over the routed tree
              + C_sw_in * (fanout taps + SB switch inputs)
  P_dyn(blk)  = 0.5 * a_blk * f_clk * Vdd^2 * C_blk(type)
  P_clock     = 0.5 * f_clk * Vdd^2 * (C_wire * clock spine length
                + C_sw_in * n_seq)          (activity 1: it's the clock)
  P_leak      = per-block leakage constants by type

Activities default to a uniform 0.15 (VPR's default static activity) or
come from an activity file of "netname activity" lines (the reference's
.act input).
"""
import numpy as np

_LEAK_W = {0: 0.2e-6, 1: 2.0e-6, 2: 8.0e-6, 3: 6.0e-6}   # W per block
_CBLK_F = {0: 2e-15, 1: 60e-15, 2: 300e-15, 3: 250e-15}  # F per block


def read_activity_file(path, netlist):
    """Per-net activity from "netname activity [prob]" lines; nets not
    named fall back to the default. Net name = driving block's name."""
    names = netlist.names or []
    by_name = {}
    with open(path) as f:
        for raw in f:
            toks = raw.split("#", 1)[0].split()
            if len(toks) >= 2:
                by_name[toks[0]] = float(toks[1])
    act = np.full(netlist.num_nets, 0.15, dtype=np.float64)
    for n in range(netlist.num_nets):
        drv = int(netlist.net_driver[n])
        if drv < len(names) and names[drv] in by_name:
            act[n] = by_name[names[drv]]
    return act


def estimate_power(netlist, arch, g, router, activities=None,
                   f_clk=100e6, vdd=0.9):
    """Returns a dict breakdown in watts. router: CPU SerialRouter (its
    route trees supply per-net wire capacitance); activities: per-net
    switching activity array or None for the 0.15 default."""
    nn = netlist.num_nets
    act = (np.asarray(activities, dtype=np.float64) if activities is not None
           else np.full(nn, 0.15))
    Cn = np.asarray(g.node_C)
    ty = np.asarray(g.type)
    half_fv2 = 0.5 * f_clk * vdd * vdd

    p_route = 0.0
    n_routed = min(nn, int(router.num_nets()))
    for n in range(n_routed):
        nodes, parents, sw, delay = router.tree(n)
        nodes = np.asarray(nodes)
        if not len(nodes):
            continue
        c_net = float(Cn[nodes].sum())
        c_net += float(arch.C_sw_in) * len(nodes)   # switch inputs along tree
        p_route += act[n] * half_fv2 * c_net

    bt = np.asarray(netlist.block_type)
    seq = np.asarray(netlist.block_is_seq).astype(bool)
    p_logic = 0.0
    p_leak = 0.0
    for t in (0, 1, 2, 3):
        cnt = int((bt == t).sum())
        p_logic += cnt * 0.15 * half_fv2 * _CBLK_F[t]
        p_leak += cnt * _LEAK_W[t]
    # clock network: H-spine across the grid + a tap per sequential block
    spine_len = arch.nx * arch.ny  # one tile-length of clock wire per tile
    c_clock = arch.C_wire * spine_len + arch.C_sw_in * int(seq.sum())
    p_clock = half_fv2 * 1.0 * c_clock   # activity 1

    total = p_route + p_logic + p_clock + p_leak
    return {"total_W": total, "routing_W": p_route, "logic_W": p_logic,
            "clock_W": p_clock, "leakage_W": p_leak,
            "f_clk_Hz": f_clk, "vdd_V": vdd}


def write_power_report(path, breakdown):
    with open(path, "w") as f:
        f.write("# power report (block-level estimate)\n")
        for k in ("total_W", "routing_W", "logic_W", "clock_W",
                  "leakage_W"):
            v = breakdown[k]
            f.write(f"{k:12s} {v * 1e3:10.4f} mW\n")
        f.write(f"f_clk {breakdown['f_clk_Hz']/1e6:.1f} MHz  "
                f"vdd {breakdown['vdd_V']:.2f} V\n")
