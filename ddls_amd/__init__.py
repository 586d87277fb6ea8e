"""ddls_amd: MI355X-native RAMP cluster simulator + PAC-ML RL partitioner.

A from-scratch rebuild of the capabilities of cwfparsonson/ddls (the PAC-ML
paper's simulator + RL framework), designed MI355X-first: flat-array simulator
state (GPU-uploadable), batched GNN policy with hand-written HIP/CDNA4 kernels,
from-scratch PPO with RCCL data-parallel gradient all-reduce over xGMI.
"""
__version__ = "0.1.0"

from .graphs import CompGraph, load_pipedream_graph
from .jobs import Job, JobQueue, JobsGenerator


def __getattr__(name):  # lazy: keep `import ddls_amd` light
    if name == "RampClusterEnvironment":
        from .cluster.environment import RampClusterEnvironment
        return RampClusterEnvironment
    if name == "RampJobPartitioningEnvironment":
        from .envs import RampJobPartitioningEnvironment
        return RampJobPartitioningEnvironment
    if name == "Action":
        from .cluster.actions import Action
        return Action
    raise AttributeError(name)


__all__ = ["CompGraph", "load_pipedream_graph", "Job", "JobQueue",
           "JobsGenerator", "RampClusterEnvironment",
           "RampJobPartitioningEnvironment", "Action", "__version__"]
