"""arch.xml subset parser.

Ingests the structural subset of VPR architecture files that the engine's
fabric model uses (reference: libarchfpga/read_xml_arch_file.c:2528
XmlReadArch): device grid hints, channel segmentation (<segmentlist>),
switch timing (<switchlist>), connection-block Fc (<fc> defaults), and the
CLB/IO pin counts from <complexblocklist>. Everything else (pb_type mode
hierarchy, interconnect detail inside clusters) is intentionally collapsed:
this engine models CLBs at the block level with equivalent pins, like the
placer/router layers of the reference do.
"""
import math
import xml.etree.ElementTree as ET

from .archdef import ArchDef


def _first_float(el, *names, default=None):
    for n in names:
        v = el.get(n)
        if v is not None:
            try:
                return float(v)
            except ValueError:
                pass
    return default


def parse_arch_xml(path_or_text, nx=None, ny=None, W=64, name=None) -> ArchDef:
    """Parse an arch.xml into an ArchDef. Grid size (nx, ny) and channel
    width W are flow inputs in VPR (auto-sized from the netlist), so they
    are parameters here with the XML supplying the fabric timing/topology.
    """
    if "\n" in str(path_or_text) or str(path_or_text).lstrip().startswith("<"):
        root = ET.fromstring(path_or_text)
        src = "<inline>"
    else:
        root = ET.parse(path_or_text).getroot()
        src = str(path_or_text)

    a = ArchDef()
    a.name = name or f"xml:{src}"
    # fixed layout from the XML (reference: <layout width= height=>);
    # explicit nx/ny arguments override, auto layouts defer to the flow
    lay = root.find("layout")
    if lay is not None:
        w_ = lay.get("width"); h_ = lay.get("height")
        if w_ is not None and h_ is not None:
            try:
                a.nx, a.ny = max(2, int(float(w_))), max(2, int(float(h_)))
                a._fixed_layout = True
            except ValueError:
                pass
    if nx:
        a.nx = nx
    if ny:
        a.ny = ny
    a.W = W if W % 2 == 0 else W + 1

    # ---- segments ----
    # Multiple <segment> entries (VTR arches declare a length
    # distribution, e.g. L1 @ freq 0.3 + L4 @ 0.7): the LONGEST length
    # becomes L and the length-1 share of the channel becomes w_l1
    # (rr_build's two-length fabric; see archdef.w_l1 for why a
    # single-length unidirectional channel cannot route locally).
    segs = root.findall(".//segmentlist/segment")
    if segs:
        def seg_len(el):
            s = el.get("length", "4")
            return 1 if s in ("longline",) else max(1, int(s))

        def seg_freq(el):
            try:
                return float(el.get("freq", el.get("frequency", "1")))
            except ValueError:
                return 1.0

        lengths = [(seg_len(el), seg_freq(el)) for el in segs]
        a.L = max(l for l, _ in lengths)
        f1 = sum(f for l, f in lengths if l == 1)
        ftot = sum(f for _, f in lengths) or 1.0
        if a.L > 1 and f1 > 0:
            a.w_l1 = max(2, int(a.W * f1 / ftot) & ~1)
        seg = max(segs, key=seg_len)   # timing from the workhorse segment
        rm = _first_float(seg, "Rmetal", default=None)
        cm = _first_float(seg, "Cmetal", default=None)
        if rm is not None:
            a.R_wire = rm
        if cm is not None:
            a.C_wire = cm

    # ---- switches ----
    sw = None
    for cand in root.findall(".//switchlist/switch"):
        sw = cand
        break
    if sw is not None:
        r = _first_float(sw, "R", default=None)
        cin = _first_float(sw, "Cin", default=None)
        tdel = _first_float(sw, "Tdel", default=None)
        if tdel is None:
            t_el = sw.find("Tdel")
            if t_el is not None:
                tdel = _first_float(t_el, "delay", default=None)
        if r is not None:
            a.R_sw = r
        if cin is not None:
            a.C_sw_in = cin
        if tdel is not None:
            a.T_sw = tdel

    # ---- complex blocks: find the CLB-like type and IO ----
    def pin_count(pb, kind):
        total = 0
        for p in pb.findall(kind):
            total += int(p.get("num_pins", "1"))
        return total

    def col_repeat(pb):
        """<gridlocations><loc type="col" repeat="N"/> (VPR 7 column
        placement of hard-block types)."""
        loc = pb.find("gridlocations/loc")
        if loc is not None and loc.get("type") == "col":
            try:
                return max(2, int(loc.get("repeat", "8")))
            except ValueError:
                pass
        return 8

    def block_delay(pb, default):
        dmx = pb.find(".//delay_constant")
        if dmx is not None:
            v = _first_float(dmx, "max", default=None)
            if v is not None:
                return v
        return default

    clb_seen = False
    for pb in root.findall(".//complexblocklist/pb_type"):
        pname = (pb.get("name") or "").lower()
        n_in = pin_count(pb, "input")
        n_out = pin_count(pb, "output")
        if pname in ("io", "inpad", "outpad"):
            cap = int(pb.get("capacity", "8"))
            a.io_cap = max(1, cap)
        elif any(k in pname for k in ("memory", "mem", "ram")):
            # RAM column hard-block type (e.g. mem32K in
            # k6_frac_N10_mem32K_40nm)
            a.ram_in = max(1, n_in)
            a.ram_out = max(1, n_out)
            a.ram_col_every = col_repeat(pb)
            a.T_ram = block_delay(pb, a.T_ram)
        elif any(k in pname for k in ("mult", "dsp")):
            a.dsp_in = max(1, n_in)
            a.dsp_out = max(1, n_out)
            a.dsp_col_every = col_repeat(pb)
            a.T_dsp = block_delay(pb, a.T_dsp)
        elif n_in > 0 and n_out > 0 and not clb_seen:
            clb_seen = True
            # cluster size: the inner BLE pb_type's num_pb
            inner = pb.find(".//pb_type[@num_pb]")
            if inner is not None:
                try:
                    a.clb_n_ble = max(1, int(inner.get("num_pb")))
                except (TypeError, ValueError):
                    pass
            # first real logic block type
            a.clb_in = n_in
            a.clb_out = n_out
            # fc defaults
            fc = pb.find("fc")
            if fc is not None:
                fin = _first_float(fc, "in_val", "default_in_val", default=None)
                fout = _first_float(fc, "out_val", "default_out_val", default=None)
                in_ty = fc.get("in_type", fc.get("default_in_type", "frac"))
                out_ty = fc.get("out_type", fc.get("default_out_type", "frac"))
                if fin is not None:
                    a.fc_in = max(1, int(round(fin * a.W)) if "frac" in in_ty
                                  else int(fin))
                if fout is not None:
                    a.fc_out = max(1, int(round(fout * a.W)) if "frac" in out_ty
                                   else int(fout))
            # sequential element timing if present
            tsu = pb.find(".//T_setup")
            if tsu is not None:
                v = _first_float(tsu, "value", default=None)
                if v is not None:
                    a.T_seq_in = v
            tcq = pb.find(".//T_clock_to_Q")
            if tcq is not None:
                v = _first_float(tcq, "max", "value", default=None)
                if v is not None:
                    a.T_seq_out = v
            a.T_clb = block_delay(pb, a.T_clb)
    a.fc_in = min(a.fc_in, a.W)
    a.fc_out = min(a.fc_out, a.W)
    return a


def size_grid_for_netlist(netlist, arch: ArchDef, fill_target=0.8):
    """VPR-style auto grid sizing (reference: SetupGrid.c): smallest square
    grid fitting every block type at fill_target (heterogeneous fabrics:
    enough RAM/DSP column tiles too), with enough IO perimeter. A fixed
    <layout width height> in the XML wins over auto sizing."""
    if getattr(arch, "_fixed_layout", False):
        return arch
    counts = {t: int((netlist.block_type == t).sum()) for t in (0, 1, 2, 3)}
    side = max(2, math.ceil(math.sqrt(max(1, counts[1]) / fill_target)))
    while True:
        arch.nx = arch.ny = side
        if 2 * (side + side) * arch.io_cap < counts[0]:
            side += 1
            continue
        ok = True
        for t in (1, 2, 3):
            if counts[t] and arch.num_tiles_of_type(t) * fill_target < counts[t]:
                ok = False
                break
        if ok:
            return arch
        side += 1
