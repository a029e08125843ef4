"""parallax_amd — MI355X-native decentralized pipeline-parallel LLM inference engine.

A from-scratch CDNA4 (gfx950) design with the capabilities of GradientHQ/parallax:
layer-range sharding across nodes, continuous batching with a paged KV cache and
block-radix prefix cache, hand-written HIP/MFMA kernels for the hot ops, RCCL over
xGMI for pipeline hidden-state transport and tensor-parallel all-reduce, and an
OpenAI-compatible HTTP frontend.
"""

__version__ = "0.1.0"
