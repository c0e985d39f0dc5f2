from tepdist_amd.planner.auto_parallel import AutoParallel, ParallelPlan  # noqa: F401
from tepdist_amd.planner.dist_spec import DimStrategy, DistSpec  # noqa: F401
from tepdist_amd.planner.cost_model import CostModel, Evaluator, HardwareModel  # noqa: F401
