"""VPR-compatible .place file writer/reader.

Format (reference: vpr/SRC/base/read_place.c print_place / read_place):
  Netlist file: <name>   Architecture file: <name>
  Array size: <nx> x <ny> logic blocks
  <blank>
  #block name	x	y	subblk	block number
  #----------	--	--	------	------------
  <name>	<x>	<y>	<subblk>	#<i>
"""


def write_place(path, placement, netlist, arch, netlist_file="netlist.net",
                arch_file="arch.xml"):
    names = netlist.names or [f"blk_{i}" for i in range(netlist.num_blocks)]
    with open(path, "w") as f:
        f.write(f"Netlist file: {netlist_file}   "
                f"Architecture file: {arch_file}\n")
        f.write(f"Array size: {arch.nx} x {arch.ny} logic blocks\n\n")
        f.write("#block name\tx\ty\tsubblk\tblock number\n")
        f.write("#----------\t--\t--\t------\t------------\n")
        for b in range(netlist.num_blocks):
            f.write(f"{names[b]}\t{placement.x[b]}\t{placement.y[b]}"
                    f"\t{placement.slot[b]}\t#{b}\n")


def read_place(path, netlist):
    """Read a .place file back; returns (x, y, slot) arrays ordered by the
    trailing block-number comment (or by name if names available)."""
    import numpy as np
    nb = netlist.num_blocks
    x = np.zeros(nb, dtype=np.int32)
    y = np.zeros(nb, dtype=np.int32)
    slot = np.zeros(nb, dtype=np.int32)
    name_to_id = None
    if netlist.names:
        name_to_id = {n: i for i, n in enumerate(netlist.names)}
    seen = 0
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line or line.startswith("#") or line.startswith("Netlist") \
                    or line.startswith("Array"):
                continue
            parts = line.split()
            if len(parts) < 4:
                continue
            if len(parts) >= 5 and parts[4].startswith("#"):
                b = int(parts[4][1:])
            elif name_to_id is not None and parts[0] in name_to_id:
                b = name_to_id[parts[0]]
            else:
                continue
            x[b] = int(parts[1]); y[b] = int(parts[2]); slot[b] = int(parts[3])
            seen += 1
    if seen != nb:
        raise ValueError(f".place file has {seen} blocks, netlist has {nb}")
    from ..place.placer import Placement
    return Placement(x, y, slot)


def read_pad_loc(path, netlist):
    """VPR pad location file (reference: -pad_loc_file, read_place.c
    read_user_pad_loc): lines of "blockname x y subblk"; '#' comments.
    Returns (ids, x, y, slot) int32 arrays for blocks found by name."""
    import numpy as np
    names = {n: i for i, n in enumerate(netlist.names or [])}
    ids, xs, ys, ss = [], [], [], []
    with open(path) as f:
        for raw in f:
            line = raw.split("#", 1)[0].strip()
            if not line:
                continue
            toks = line.split()
            if len(toks) < 3:
                raise ValueError(f"bad pad_loc line: {raw!r}")
            name = toks[0]
            if name not in names:
                raise ValueError(f"pad_loc: unknown block {name!r}")
            ids.append(names[name])
            xs.append(int(toks[1]))
            ys.append(int(toks[2]))
            ss.append(int(toks[3]) if len(toks) > 3 else 0)
    return (np.asarray(ids, dtype=np.int32), np.asarray(xs, dtype=np.int32),
            np.asarray(ys, dtype=np.int32), np.asarray(ss, dtype=np.int32))
