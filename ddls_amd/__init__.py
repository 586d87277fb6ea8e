"""ddls_amd: MI355X-native RAMP cluster simulator + PAC-ML RL partitioner.

A from-scratch rebuild of the capabilities of cwfparsonson/ddls (the PAC-ML
paper's simulator + RL framework), designed MI355X-first: flat-array simulator
state (GPU-uploadable), batched GNN policy with hand-written HIP/CDNA4 kernels,
from-scratch PPO with RCCL data-parallel gradient all-reduce over xGMI.
"""
__version__ = "0.1.0"

from .graphs import CompGraph, load_pipedream_graph
from .jobs import Job, JobQueue, JobsGenerator

__all__ = ["CompGraph", "load_pipedream_graph", "Job", "JobQueue",
           "JobsGenerator", "__version__"]
