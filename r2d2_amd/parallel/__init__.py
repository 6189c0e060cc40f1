from .ddp import GradAllReducer  # noqa: F401
