from .taskflow import TASKS, Taskflow  # noqa: F401
