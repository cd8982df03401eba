"""MI355X-native inference gateway.

A from-scratch rebuild of the capabilities of inference-gateway/inference-gateway
 around an in-process PyTorch-ROCm + hand-written CDNA4 HIP
kernel inference engine for AMD Instinct MI355X.
"""

__version__ = "0.1.0"
