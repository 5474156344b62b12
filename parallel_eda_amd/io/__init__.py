from .synth import synth_netlist, synth_placed_netlist, SynthSpec, NetlistPy

__all__ = ["synth_netlist", "synth_placed_netlist", "SynthSpec", "NetlistPy"]
