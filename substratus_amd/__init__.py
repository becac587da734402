"""substratus_amd — an MI355X-native ML platform with the capabilities of
substratusai/substratus.

Two halves:
  * Control plane: `substratus.ai/v1` CRD API (Model/Dataset/Server/Notebook),
    controllers, cloud abstraction, SCI, `sub` CLI — reference internal/{controller,
    cloud,sci,cli} (Go); re-implemented in Python against a pluggable Kubernetes
    API (in-memory for tests, real cluster when available).
  * Data plane: first-party loader/trainer/server replacing the reference's
    external HuggingFace/Basaran images — PyTorch-ROCm + hand-written CDNA4 HIP
    kernels (substratus_amd/ops/hip) + RCCL over xGMI.
"""

__version__ = "0.1.0"
