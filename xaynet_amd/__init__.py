"""xaynet_amd — MI355X-native masked federated learning (PET protocol).

A from-scratch rebuild of xaynetwork/xaynet's capabilities for AMD MI355X:
the coordinator's masked-aggregation data plane runs as hand-written
HIP/CDNA4 kernels with the model vector sharded over RCCL/xGMI; the
protocol core (crypto, masking, wire formats) is native C++ with Python
bindings API-compatible with the reference's `xaynet_sdk` package.
"""

__version__ = "0.1.0"
