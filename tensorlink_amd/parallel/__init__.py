from tensorlink_amd.parallel.planner import ModelParser, StagePlan, StageSpec, AssignmentError  # noqa: F401
