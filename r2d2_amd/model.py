"""Reference-parity import site: ``from r2d2_amd.model import Network, AgentState``
(reference: /root/reference/model.py)."""

from .models.network import Network, AgentState  # noqa: F401
