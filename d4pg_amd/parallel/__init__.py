from .worker import Worker, global_model_eval  # noqa: F401
