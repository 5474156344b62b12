from .synth import synth_netlist, SynthSpec

__all__ = ["synth_netlist", "SynthSpec"]
