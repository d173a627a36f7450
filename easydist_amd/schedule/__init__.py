from .efficient_memory_scheduler import plan_memory  # noqa: F401
