"""vizier_amd: an MI355X-native black-box optimization service.

A from-scratch implementation of OSS Vizier's capabilities — pyvizier
data model, wire-compatible gRPC study service, Pythia policy protocol,
designer/benchmark libraries — whose GP-Bandit numeric core runs on
hand-written CDNA4 (gfx950) HIP kernels over PyTorch-ROCm, with
RCCL/xGMI data-parallel acquisition sweeps.

Public entry points:
  from vizier_amd import pyvizier as vz
  from vizier_amd.service import clients
  from vizier_amd import pythia, algorithms, benchmarks
"""

__version__ = '0.1.0'
