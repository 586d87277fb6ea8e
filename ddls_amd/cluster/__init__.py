from .environment import RampClusterEnvironment
from .actions import (Action, OpPartition, OpPlacement, OpSchedule,
                      DepPlacement, DepSchedule)

__all__ = ["RampClusterEnvironment", "Action", "OpPartition", "OpPlacement",
           "OpSchedule", "DepPlacement", "DepSchedule"]
