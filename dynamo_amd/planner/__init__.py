from .planner import (ARIMAPredictor, Connector, ConstantPredictor,
                      CorrectionFactors, InterpolatedPerfModel, LoadPlanner,
                      MovingAveragePredictor, PerfModel, PlannerService,
                      PoolObservation, PoolPolicy, ScalingState, SLAPlanner,
                      SLATargets, ThroughputPlanner, TrendPredictor,
                      VirtualConnector)

__all__ = ["ARIMAPredictor", "Connector", "ConstantPredictor",
           "CorrectionFactors", "InterpolatedPerfModel", "LoadPlanner",
           "MovingAveragePredictor", "PerfModel", "PlannerService",
           "PoolObservation", "PoolPolicy", "ScalingState", "SLAPlanner",
           "SLATargets", "ThroughputPlanner", "TrendPredictor",
           "VirtualConnector"]
