"""Post-route structural Verilog + SDF writers.

Reference: base/verilog_writer.c (post-synthesis netlist + SDF timing
annotation). Emits the block-level structure: IO ports, one module
instance per logic block (CLB/RAM/DSP), wires per net — enough for
downstream structural consumers / equivalence checks at the cluster
level (intra-cluster logic is behavioral in our model). write_sdf emits
the matching SDF 2.1 file: IOPATH delays per block and INTERCONNECT
delays per routed connection.
"""
from ..arch.archdef import BLK_NAMES


def _sanitize(n):
    return "".join(c if c.isalnum() or c == "_" else "_" for c in n)


def write_verilog(path, netlist, design="top"):
    names = netlist.names or [f"blk_{i}" for i in range(netlist.num_blocks)]
    sanitize = _sanitize
    nd = netlist.net_driver
    sp = netlist.net_sink_ptr
    ss = netlist.net_sinks
    bt = netlist.block_type
    in_pads = [b for b in range(netlist.num_blocks)
               if bt[b] == 0 and (nd == b).any()]
    out_pads = [b for b in range(netlist.num_blocks)
                if bt[b] == 0 and b not in set(in_pads)]
    net_name = [f"n_{sanitize(names[nd[n]])}" for n in range(netlist.num_nets)]

    with open(path, "w") as f:
        ports = [f"input {sanitize(names[b])}" for b in in_pads] + \
                [f"output {sanitize(names[b])}" for b in out_pads]
        f.write(f"module {sanitize(design)} (\n  " + ",\n  ".join(ports) +
                "\n);\n\n")
        for n in range(netlist.num_nets):
            f.write(f"  wire {net_name[n]};\n")
        f.write("\n")
        # input pads drive their nets
        for b in in_pads:
            for n in range(netlist.num_nets):
                if nd[n] == b:
                    f.write(f"  assign {net_name[n]} = {sanitize(names[b])};\n")
        # output pads driven by the net that sinks into them
        for b in out_pads:
            for n in range(netlist.num_nets):
                if (ss[sp[n]:sp[n + 1]] == b).any():
                    f.write(f"  assign {sanitize(names[b])} = {net_name[n]};\n")
        f.write("\n")
        # logic block instances (CLB / RAM / DSP)
        for b in range(netlist.num_blocks):
            if bt[b] == 0:
                continue
            ins = [net_name[n] for n in range(netlist.num_nets)
                   if (ss[sp[n]:sp[n + 1]] == b).any()]
            outs = [net_name[n] for n in range(netlist.num_nets) if nd[n] == b]
            base = BLK_NAMES[bt[b]]
            kind = f"{base}_seq" if netlist.block_is_seq[b] else f"{base}_comb"
            conns = [f".i{k}({w})" for k, w in enumerate(ins)] + \
                    [f".o{k}({w})" for k, w in enumerate(outs)]
            f.write(f"  {kind} {sanitize(names[b])} (" + ", ".join(conns)
                    + ");\n")
        f.write("\nendmodule\n")


def write_sdf(path, netlist, arch, conn_delay, design="top"):
    """SDF 2.1 timing annotation for the structural netlist (reference:
    verilog_writer.c's SDF output): one CELL per logic block with an
    IOPATH of its combinational (or clock-to-Q) delay, plus one
    fpga_interconnect CELL carrying an INTERCONNECT entry per routed
    connection. conn_delay: seconds per (net,sink) connection, aligned
    with netlist.net_sinks (routed delays from the router, or placement
    estimates)."""
    names = netlist.names or [f"blk_{i}" for i in range(netlist.num_blocks)]
    nd = netlist.net_driver
    sp = netlist.net_sink_ptr
    ss = netlist.net_sinks
    bt = netlist.block_type

    def ps(seconds):
        v = seconds * 1e12
        return f"({v:.1f}:{v:.1f}:{v:.1f})"

    with open(path, "w") as f:
        f.write("(DELAYFILE\n")
        f.write('  (SDFVERSION "2.1")\n')
        f.write(f'  (DESIGN "{_sanitize(design)}")\n')
        f.write("  (TIMESCALE 1 ps)\n")
        for b in range(netlist.num_blocks):
            if bt[b] == 0:
                continue
            base = BLK_NAMES[bt[b]]
            if netlist.block_is_seq[b]:
                d = arch.T_seq_out
                pathspec = f"(IOPATH (posedge clk) O {ps(d)} {ps(d)})"
            else:
                d = arch.block_delay_of(int(bt[b]))
                pathspec = f"(IOPATH I O {ps(d)} {ps(d)})"
            f.write(f'  (CELL (CELLTYPE "{base}")\n'
                    f"    (INSTANCE {_sanitize(names[b])})\n"
                    f"    (DELAY (ABSOLUTE {pathspec}))\n  )\n")
        f.write('  (CELL (CELLTYPE "fpga_interconnect")\n'
                "    (INSTANCE routing)\n    (DELAY (ABSOLUTE\n")
        for n in range(netlist.num_nets):
            for c in range(sp[n], sp[n + 1]):
                d = float(conn_delay[c])
                f.write(f"      (INTERCONNECT {_sanitize(names[nd[n]])}/O "
                        f"{_sanitize(names[ss[c]])}/I {ps(d)} {ps(d)})\n")
        f.write("    ))\n  )\n)\n")
