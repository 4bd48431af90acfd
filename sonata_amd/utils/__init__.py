from .trace import StageTimes, stage_timer, get_stage_times  # noqa: F401
