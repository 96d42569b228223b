from .planner import (Connector, ConstantPredictor, LoadPlanner,
                      MovingAveragePredictor, PerfModel, PlannerService,
                      PoolObservation, PoolPolicy, SLATargets,
                      ThroughputPlanner, TrendPredictor, VirtualConnector)

__all__ = ["Connector", "ConstantPredictor", "LoadPlanner",
           "MovingAveragePredictor", "PerfModel", "PlannerService",
           "PoolObservation", "PoolPolicy", "SLATargets",
           "ThroughputPlanner", "TrendPredictor", "VirtualConnector"]
