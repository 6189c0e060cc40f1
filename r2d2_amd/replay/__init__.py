from .priority_tree import PriorityTree  # noqa: F401
