"""Packed-netlist (.net) writer/reader.

The reference reads/writes an XML .net (output_clustering.c:609 writer,
read_netlist.c reader). We keep the same XML shape at the block level:
<block name instance mode> with <inputs>/<outputs> port connections, so a
netlist round-trips through the file and external tools can inspect it.
Intra-cluster detail is flattened (the engine models clusters atomically).
"""
import xml.etree.ElementTree as ET

import numpy as np

from .synth import NetlistPy
from ..arch.archdef import BLK_IO, BLK_CLB, BLK_NAMES


def write_net(path, netlist: NetlistPy, design_name="top"):
    names = netlist.names or [f"blk_{i}" for i in range(netlist.num_blocks)]
    root = ET.Element("block", name=design_name, instance="FPGA_packed_netlist")
    # net names: by driving block
    net_name = [f"net_{names[netlist.net_driver[n]]}"
                for n in range(netlist.num_nets)]
    # per block: incoming nets (inputs), outgoing nets (outputs)
    in_nets = [[] for _ in range(netlist.num_blocks)]
    out_nets = [[] for _ in range(netlist.num_blocks)]
    for n in range(netlist.num_nets):
        out_nets[netlist.net_driver[n]].append(n)
        for s in netlist.net_sinks[netlist.net_sink_ptr[n]:netlist.net_sink_ptr[n + 1]]:
            in_nets[s].append(n)
    for b in range(netlist.num_blocks):
        ty = BLK_NAMES[netlist.block_type[b]]
        el = ET.SubElement(root, "block", name=names[b],
                           instance=f"{ty}[{b}]",
                           mode="seq" if netlist.block_is_seq[b] else "comb")
        ins = ET.SubElement(el, "inputs")
        ins.text = " ".join(net_name[n] for n in in_nets[b]) or "open"
        outs = ET.SubElement(el, "outputs")
        outs.text = " ".join(net_name[n] for n in out_nets[b]) or "open"
    ET.indent(ET.ElementTree(root))
    ET.ElementTree(root).write(path)


def read_net(path) -> NetlistPy:
    root = ET.parse(path).getroot()
    names, types, seqs = [], [], []
    in_lists, out_lists = [], []
    for el in root.findall("block"):
        names.append(el.get("name"))
        inst = el.get("instance", "clb[0]")
        kind = inst.split("[", 1)[0]
        types.append(BLK_NAMES.index(kind) if kind in BLK_NAMES else BLK_CLB)
        seqs.append(1 if el.get("mode") == "seq" else 0)
        ins = (el.findtext("inputs") or "").split()
        outs = (el.findtext("outputs") or "").split()
        in_lists.append([s for s in ins if s != "open"])
        out_lists.append([s for s in outs if s != "open"])
    nb = len(names)
    # rebuild nets: each distinct net name has one driver + sinks
    driver_of = {}
    for b in range(nb):
        for nn in out_lists[b]:
            driver_of[nn] = b
    net_names = sorted(driver_of)
    net_id = {nn: i for i, nn in enumerate(net_names)}
    sink_lists = [[] for _ in net_names]
    for b in range(nb):
        for nn in in_lists[b]:
            if nn in net_id:
                sink_lists[net_id[nn]].append(b)
    drivers = np.asarray([driver_of[nn] for nn in net_names], dtype=np.int32)
    sink_ptr = np.zeros(len(net_names) + 1, dtype=np.int64)
    for i, s in enumerate(sink_lists):
        sink_ptr[i + 1] = sink_ptr[i] + len(s)
    sinks = (np.concatenate([np.asarray(s, dtype=np.int32)
                             for s in sink_lists])
             if any(sink_lists) else np.zeros(0, dtype=np.int32))
    # drop sinkless nets
    keep = np.diff(sink_ptr) > 0
    if not keep.all():
        kept = np.nonzero(keep)[0]
        new_ptr = np.zeros(len(kept) + 1, dtype=np.int64)
        segs = []
        for i, k in enumerate(kept):
            seg = sinks[sink_ptr[k]:sink_ptr[k + 1]]
            segs.append(seg)
            new_ptr[i + 1] = new_ptr[i] + len(seg)
        drivers = drivers[kept]
        sink_ptr = new_ptr
        sinks = np.concatenate(segs) if segs else np.zeros(0, dtype=np.int32)
    return NetlistPy(np.asarray(types, dtype=np.int8),
                     np.asarray(seqs, dtype=np.uint8),
                     drivers, sink_ptr, sinks, names=names)


def check_netlist(netlist: NetlistPy):
    """Netlist consistency validator (reference: base/check_netlist.c:37)."""
    errs = []
    nb = netlist.num_blocks
    if (netlist.net_driver < 0).any() or (netlist.net_driver >= nb).any():
        errs.append("net driver out of range")
    if len(netlist.net_sinks) and ((netlist.net_sinks < 0).any() or
                                   (netlist.net_sinks >= nb).any()):
        errs.append("net sink out of range")
    for n in range(netlist.num_nets):
        s = netlist.net_sinks[netlist.net_sink_ptr[n]:netlist.net_sink_ptr[n + 1]]
        if len(s) == 0:
            errs.append(f"net {n} has no sinks")
        if (s == netlist.net_driver[n]).any():
            errs.append(f"net {n} drives itself")
        if len(np.unique(s)) != len(s):
            errs.append(f"net {n} has duplicate sinks")
    # every non-driving, non-sinking block is dangling
    used = np.zeros(nb, dtype=bool)
    used[netlist.net_driver] = True
    used[netlist.net_sinks] = True
    dangling = int((~used).sum())
    return errs, dangling
