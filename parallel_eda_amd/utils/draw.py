"""Headless placement/routing rendering to SVG.

Reference scope: base/graphics.c + draw.c drive an interactive X11 view
of the placement and routing; SURVEY §7 plans the headless equivalent —
dump-to-image. One SVG per call, no external dependencies.
"""
from ..arch.archdef import BLK_NAMES

_FILL = {0: "#9ecae1", 1: "#c7e9c0", 2: "#fdae6b", 3: "#bcbddc"}


def _header(w, h, scale):
    return (f'<svg xmlns="http://www.w3.org/2000/svg" '
            f'width="{w * scale}" height="{h * scale}" '
            f'viewBox="0 0 {w * scale} {h * scale}">\n'
            f'<rect width="100%" height="100%" fill="white"/>\n')


def write_placement_svg(path, placement, netlist, arch, scale=14):
    """Tile grid colored by block type, one cell per occupied slot."""
    gx, gy = arch.nx + 2, arch.ny + 2
    s = scale
    out = [_header(gx, gy, s)]
    # tile outlines (logic area)
    for x in range(1, arch.nx + 1):
        t = arch.col_block_type(x)
        for y in range(1, arch.ny + 1):
            out.append(f'<rect x="{x*s}" y="{(gy-1-y)*s}" width="{s}" '
                       f'height="{s}" fill="none" stroke="#eee"/>')
            if t != 1:
                out.append(f'<rect x="{x*s}" y="{(gy-1-y)*s}" width="{s}" '
                           f'height="{s}" fill="{_FILL[t]}" '
                           f'fill-opacity="0.25"/>')
    bt = netlist.block_type
    for b in range(netlist.num_blocks):
        x, y = int(placement.x[b]), int(placement.y[b])
        out.append(f'<rect x="{x*s+1}" y="{(gy-1-y)*s+1}" width="{s-2}" '
                   f'height="{s-2}" fill="{_FILL[int(bt[b])]}">'
                   f'<title>blk {b} ({BLK_NAMES[int(bt[b])]}) '
                   f'@({x},{y})</title></rect>')
    out.append("</svg>\n")
    with open(path, "w") as f:
        f.write("\n".join(out))


def write_routing_svg(path, g, arch, router, net_ids=None, scale=14,
                      max_nets=2000):
    """Routed wires as segments; overused nodes highlighted red."""
    import numpy as np
    gx, gy = arch.nx + 2, arch.ny + 2
    s = scale
    ty = np.asarray(g.type)
    xl = np.asarray(g.xlow); yl = np.asarray(g.ylow)
    xh = np.asarray(g.xhigh); yh = np.asarray(g.yhigh)
    cap = np.asarray(g.capacity)
    occ = np.asarray(router.occ())
    out = [_header(gx, gy, s)]
    n = 0
    colors = ["#3182bd", "#31a354", "#756bb1", "#636363", "#e6550d"]
    limit = (len(net_ids) if net_ids is not None
             else int(router.num_nets()))
    while n < limit:
        nodes, parents, sw, delay = router.tree(n)
        col = colors[n % len(colors)]
        for v in np.asarray(nodes):
            if ty[v] not in (4, 5):
                continue
            x0, y0 = xl[v], yl[v]
            x1, y1 = xh[v], yh[v]
            over = occ[v] > cap[v]
            c = "#de2d26" if over else col
            w = 2.5 if over else 1.0
            out.append(
                f'<line x1="{x0*s+s//2}" y1="{(gy-1-y0)*s+s//2}" '
                f'x2="{x1*s+s//2}" y2="{(gy-1-y1)*s+s//2}" '
                f'stroke="{c}" stroke-width="{w}" stroke-opacity="0.6"/>')
        n += 1
        if n >= max_nets:
            break
    out.append("</svg>\n")
    with open(path, "w") as f:
        f.write("\n".join(out))
