"""BLIF reader (technology-mapped netlists).

Re-implements the semantics of the reference's read_blif.c (1,981 LoC):
parses .model/.inputs/.outputs/.names/.latch, sweeps dangling nets, and
produces a primitive-level netlist (LUTs + latches + IO pads) ready for
packing (pack.py) into the block-level netlist the placer/router consume.
.subckt is accepted only for simple single-output cells.
"""
from dataclasses import dataclass, field


@dataclass
class BlifPrimitive:
    kind: str               # "input", "output", "names", "latch"
    name: str               # output signal name (or pad signal)
    inputs: list = field(default_factory=list)
    clock: str = ""


@dataclass
class BlifModel:
    name: str = ""
    inputs: list = field(default_factory=list)
    outputs: list = field(default_factory=list)
    prims: list = field(default_factory=list)     # names/latch primitives

    def signal_drivers(self):
        drv = {}
        for s in self.inputs:
            drv[s] = ("input", s)
        for p in self.prims:
            drv[p.name] = (p.kind, p.name)
        return drv


def _tokens_of_lines(text):
    """BLIF line continuation (backslash) + comment stripping."""
    logical = []
    pending = ""
    for raw in text.splitlines():
        line = raw.split("#", 1)[0].rstrip()
        if not line.strip() and not pending:
            continue
        if line.endswith("\\"):
            pending += line[:-1] + " "
            continue
        logical.append((pending + line).strip())
        pending = ""
    if pending.strip():
        logical.append(pending.strip())
    return logical


def parse_blif(text) -> BlifModel:
    model = BlifModel()
    lines = _tokens_of_lines(text)
    i = 0
    n = len(lines)
    while i < n:
        toks = lines[i].split()
        i += 1
        if not toks:
            continue
        key = toks[0]
        if key == ".model":
            if model.name:
                break  # only the first (top) model is read
            model.name = toks[1] if len(toks) > 1 else "top"
        elif key == ".inputs":
            model.inputs.extend(toks[1:])
        elif key == ".outputs":
            model.outputs.extend(toks[1:])
        elif key == ".names":
            sigs = toks[1:]
            out = sigs[-1]
            ins = sigs[:-1]
            # skip the cover rows
            while i < n and not lines[i].startswith("."):
                i += 1
            model.prims.append(BlifPrimitive("names", out, ins))
        elif key == ".latch":
            # .latch input output [type clock] [init]
            inp, out = toks[1], toks[2]
            clock = toks[4] if len(toks) > 4 else ""
            model.prims.append(BlifPrimitive("latch", out, [inp], clock))
        elif key == ".subckt":
            # minimal support: treat as a comb primitive with the last
            # formal=actual as output
            conns = [t.split("=") for t in toks[2:] if "=" in t]
            if conns:
                out = conns[-1][1]
                ins = [c[1] for c in conns[:-1]]
                model.prims.append(BlifPrimitive("names", out, ins))
        elif key == ".end":
            break
    sweep(model)
    return model


def sweep(model: BlifModel):
    """Remove primitives whose outputs drive nothing (dangling), like the
    reference's sweep of hanging nets."""
    used = set(model.outputs)
    for p in model.prims:
        used.update(p.inputs)
        if p.clock:
            used.add(p.clock)
    model.prims = [p for p in model.prims
                   if p.name in used or p.name in model.outputs]


def read_blif(path) -> BlifModel:
    with open(path) as f:
        return parse_blif(f.read())
