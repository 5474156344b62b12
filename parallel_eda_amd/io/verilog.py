"""Post-route structural Verilog writer.

Reference: base/verilog_writer.c (post-synthesis netlist). Emits the
block-level structure: IO ports, one module instance per CLB, wires per
net — enough for downstream structural consumers / equivalence checks at
the cluster level (intra-cluster logic is behavioral in our model).
"""


def write_verilog(path, netlist, design="top"):
    names = netlist.names or [f"blk_{i}" for i in range(netlist.num_blocks)]

    def sanitize(n):
        return "".join(c if c.isalnum() or c == "_" else "_" for c in n)

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
        # CLB instances
        for b in range(netlist.num_blocks):
            if bt[b] != 1:
                continue
            ins = [net_name[n] for n in range(netlist.num_nets)
                   if (ss[sp[n]:sp[n + 1]] == b).any()]
            outs = [net_name[n] for n in range(netlist.num_nets) if nd[n] == b]
            kind = "clb_seq" if netlist.block_is_seq[b] else "clb_comb"
            conns = [f".i{k}({w})" for k, w in enumerate(ins)] + \
                    [f".o{k}({w})" for k, w in enumerate(outs)]
            f.write(f"  {kind} {sanitize(names[b])} (" + ", ".join(conns)
                    + ");\n")
        f.write("\nendmodule\n")
