from .model import LoadedModel, ModelExecError, load_model_from_dir  # noqa: F401
from .planner import Plan, PlanError, compile_graph  # noqa: F401
