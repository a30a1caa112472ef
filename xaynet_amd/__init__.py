"""xaynet_amd — MI355X-native masked federated learning (PET protocol).

A from-scratch rebuild of xaynetwork/xaynet's capabilities for AMD MI355X:
the coordinator's masked-aggregation data plane runs as hand-written
HIP/CDNA4 kernels with digit-plane accumulators reduced over RCCL/xGMI; the
protocol core (crypto, masking, wire formats, state machines) is native C++
with Python bindings API-compatible with the reference's `xaynet_sdk`
package (see the top-level `xaynet_sdk/` package in this repo).

Quick access:
    from xaynet_amd import _core          # protocol core + coordinator + SDK
    from xaynet_amd.server import Settings, serve   # coordinator daemon
    from xaynet_amd.ops import GpuMaskedAggregator  # GPU engine (needs torch)
"""

__version__ = "0.1.0"
