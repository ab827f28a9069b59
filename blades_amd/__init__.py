"""blades_amd — MI355X-native Byzantine-robust federated-learning framework.

A from-scratch framework with the capabilities and API of bladesteam/blades
(the reference simulator), redesigned for AMD Instinct MI355X: one process
per GPU over RCCL/xGMI, fused many-model client training in HBM3E-resident
slabs, and hand-written CDNA4 HIP kernels (MFMA/LDS) for the robust
aggregation and attack hot path.  See SURVEY.md for the layer map.
"""

__version__ = "0.1.0"

from blades_amd.client import BladesClient, ByzantineClient
from blades_amd.server import BladesServer
from blades_amd.simulator import Simulator

__all__ = ["Simulator", "BladesClient", "ByzantineClient", "BladesServer",
           "__version__"]
