"""bflc_amd — MI355X-native committee-consensus federated learning.

A ground-up re-design of the capabilities of iammcy/BFLC-demo for one
8x MI355X node: deterministic C++ committee ledger replicated per rank,
RCCL collectives over xGMI as the ordering/transport substrate, and
hand-written CDNA4 (gfx950) HIP kernels for the FL hot path.
See DESIGN.md and SURVEY.md.
"""
from bflc_amd.config import FLConfig

__version__ = "0.1.0"
__all__ = ["FLConfig"]
