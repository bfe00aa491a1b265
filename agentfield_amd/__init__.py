"""agentfield_amd — an MI355X-native agent control plane + in-process
PyTorch-ROCm inference engine with hand-written CDNA4 (gfx950) HIP kernels.

Capability surface mirrors Agent-Field/agentfield; the
app.ai() seam is served by the local engine instead of external providers.
"""
__version__ = "0.1.0"
