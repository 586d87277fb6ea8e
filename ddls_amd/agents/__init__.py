from .partitioners import SipMlOpPartitioner, RandomOpPartitioner
from .placers import RampFirstFitOpPlacer, RandomOpPlacer, FirstFitDepPlacer
from .schedulers import SRPTOpScheduler, SRPTDepScheduler

__all__ = ["SipMlOpPartitioner", "RandomOpPartitioner", "RampFirstFitOpPlacer",
           "RandomOpPlacer", "FirstFitDepPlacer", "SRPTOpScheduler",
           "SRPTDepScheduler"]
