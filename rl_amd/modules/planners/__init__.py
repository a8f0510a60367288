from .planners import CEMPlanner, MPCPlannerBase, MPPIPlanner
