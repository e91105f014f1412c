"""helix_amd — MI355X-native private GenAI agent stack.

A brand-new framework with the capabilities of helixml/helix, built
MI355X-first: hand-written CDNA4 (gfx950) HIP kernels for the inference
hot path, an HBM-aware model scheduler, RCCL-over-xGMI tensor
parallelism, and an OpenAI-compatible control plane with helix.yaml
agents (see SURVEY.md for the reference feature map).
"""

__version__ = "0.1.0"
