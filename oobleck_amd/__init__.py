# oobleck_amd — MI355X-native rebuild of Oobleck's pipeline-parallel
# execution hot path: hand-written CDNA4 HIP kernels behind a
# C-ABI (include/oobleck_stage.h), driven by a Python host that mirrors the
# reference's OobleckPipeline / Layer / DataParallelEngine surfaces so the
# planner and elastic logic drop in unchanged.
from .config import ModelConfig, TrainingConfig  # noqa: F401
