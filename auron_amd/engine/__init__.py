from .executor import ExecContext, Executor  # noqa: F401
