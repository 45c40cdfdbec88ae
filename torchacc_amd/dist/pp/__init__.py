from .pipeline import PipelineParallel  # noqa: F401
