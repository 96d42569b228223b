from .global_planner import (GlobalPlanner, GlobalPlannerConnector,
                             GlobalPlannerService, PoolBudgetPolicy)
from .planner import (ARIMAPredictor, Connector, ConstantPredictor,
                      CorrectionFactors, InterpolatedPerfModel, LoadPlanner,
                      MovingAveragePredictor, PerfModel, PlannerService,
                      PoolObservation, PoolPolicy, ScalingState, SLAPlanner,
                      SLATargets, ThroughputPlanner, TrendPredictor,
                      VirtualConnector)

__all__ = ["ARIMAPredictor", "GlobalPlanner", "GlobalPlannerConnector",
           "GlobalPlannerService", "PoolBudgetPolicy", "Connector", "ConstantPredictor",
           "CorrectionFactors", "InterpolatedPerfModel", "LoadPlanner",
           "MovingAveragePredictor", "PerfModel", "PlannerService",
           "PoolObservation", "PoolPolicy", "ScalingState", "SLAPlanner",
           "SLATargets", "ThroughputPlanner", "TrendPredictor",
           "VirtualConnector"]
