from .ramp_job_partitioning import RampJobPartitioningEnvironment
from .observation import RampJobPartitioningObservation
from .actors import ACTORS

__all__ = ["RampJobPartitioningEnvironment", "RampJobPartitioningObservation",
           "ACTORS"]
