"""fl4health_amd — MI355X-native federated learning framework.

A from-scratch re-design of the capabilities of VectorInstitute/FL4Health for
AMD Instinct MI355X (gfx950): PyTorch-ROCm client training, hand-written
CDNA4 HIP kernels for the per-step and per-round hot ops, and RCCL
collectives over xGMI (one client process per GPU) in place of the
reference's Flower gRPC NumPy round-trip.
"""

__version__ = "0.2.0"
