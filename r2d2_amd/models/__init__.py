from .network import Network, AgentState  # noqa: F401
from .encoders import NatureCNN, MLPEncoder, ImpalaCNN, make_encoder  # noqa: F401
