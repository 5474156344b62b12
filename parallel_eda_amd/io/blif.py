"""BLIF reader (technology-mapped netlists).

Re-implements the semantics of the reference's read_blif.c (1,981 LoC):
parses .model/.inputs/.outputs/.names/.latch/.subckt, sweeps dangling
nets, and produces a primitive-level netlist (LUTs + latches + hard
blocks + IO pads) ready for packing (pack.py) into the block-level
netlist the placer/router consume.

.subckt instances are hard blocks (VTR convention: single_port_ram /
dual_port_ram / multiply / adder ...): kept as one multi-output
primitive each and mapped by the packer to a RAM or DSP block on a
heterogeneous fabric (arch/archdef.py column tiles).
"""
from dataclasses import dataclass, field


@dataclass
class BlifPrimitive:
    kind: str               # "input", "output", "names", "latch", "subckt"
    name: str               # output signal name (or pad signal)
    inputs: list = field(default_factory=list)
    clock: str = ""
    model: str = ""         # subckt model name
    outputs: list = field(default_factory=list)  # subckt: all output sigs


# output-formal prefixes of the VTR primitive models (single_port_ram:
# "out"; dual_port_ram: "out1"/"out2"; multiply/adder: "out"/"cout"/"sumout")
_SUBCKT_OUT_PREFIXES = ("out", "q", "dataout", "spo", "dpo", "sum", "cout")


def subckt_class(model_name: str) -> str:
    """Classify a subckt model: "ram", "dsp" or "soft" (unknown cell)."""
    n = model_name.lower()
    if "ram" in n or "mem" in n or "rom" in n:
        return "ram"
    if "mult" in n or "mac" in n or "dsp" in n or "add" in n:
        return "dsp"
    return "soft"


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
            for o in (p.outputs if p.kind == "subckt" else [p.name]):
                drv[o] = (p.kind, o)
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
            # .subckt model formal=actual ... — split formals into
            # inputs/outputs by the VTR model conventions; clock formals
            # ("clk"/"clock") are global like latch clocks
            mdl = toks[1] if len(toks) > 1 else "cell"
            conns = [t.split("=", 1) for t in toks[2:] if "=" in t]
            ins, outs, clock = [], [], ""
            for f, a in conns:
                fl = f.lower()
                if fl in ("clk", "clock"):
                    clock = a
                elif fl.startswith(_SUBCKT_OUT_PREFIXES):
                    outs.append(a)
                else:
                    ins.append(a)
            if not outs and conns:  # unknown cell: last formal is output
                outs = [conns[-1][1]]
                ins = [a for _, a in conns[:-1]]
            if outs:
                model.prims.append(BlifPrimitive(
                    "subckt", outs[0], ins, clock, mdl, outs))
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

    def live(p):
        outs = p.outputs if p.kind == "subckt" else [p.name]
        return any(o in used for o in outs)

    model.prims = [p for p in model.prims if live(p)]


def read_blif(path) -> BlifModel:
    with open(path) as f:
        return parse_blif(f.read())
