"""The R2D2 agent network: encoder -> LSTM -> dueling double-Q head.

Golden eager implementation.  API mirrors the reference Network
(/root/reference/model.py:27-150): ``forward(AgentState)`` for single-step
actor inference, ``calculate_q_`` for full-sequence target-position Q, and
``calculate_q`` for learning-position online Q — plus ``calculate_q_both``,
a fused single-pass variant used by the MI355X learner (one forward yields
both the learning-position Q and the target-position Q, saving one of the
reference's three per-update network passes, same numerics).

The ragged-sequence gathers are vectorized index arithmetic rather than the
reference's per-sample Python loops (model.py:102-111,143).
"""

from dataclasses import dataclass, field
from typing import Optional, Tuple

import torch
import torch.nn as nn
from torch.nn.utils.rnn import pack_padded_sequence, pad_packed_sequence

from .encoders import make_encoder


@dataclass
class AgentState:
    """Per-env inference state (batched: leading dim = num parallel envs).

    Reference: model.py:9-24.  The reference's class-level shared-tensor
    default for last_reward (model.py:14) is deliberately NOT inherited —
    fields use default factories.
    """
    obs: torch.Tensor                    # (B, *obs_shape) uint8/float
    action_dim: int
    last_action: torch.Tensor = field(init=False)      # (B, A) float one-hot
    last_reward: torch.Tensor = field(init=False)      # (B, 1) float
    hidden_state: Optional[Tuple[torch.Tensor, torch.Tensor]] = None

    def __post_init__(self):
        b = self.obs.shape[0]
        self.last_action = torch.zeros((b, self.action_dim), dtype=torch.float32)
        self.last_action[:, 0] = 1.0
        self.last_reward = torch.zeros((b, 1), dtype=torch.float32)

    def update(self, obs, last_action, last_reward, hidden):
        """Advance the state after an env step.  obs: np/tensor (B, *shape);
        last_action: (B,) int; last_reward: (B,) float."""
        if not torch.is_tensor(obs):
            obs = torch.from_numpy(obs)
        self.obs = obs
        la = torch.as_tensor(last_action, dtype=torch.long).view(-1)
        self.last_action = torch.zeros((la.numel(), self.action_dim), dtype=torch.float32)
        self.last_action[torch.arange(la.numel()), la] = 1.0
        self.last_reward = torch.as_tensor(last_reward, dtype=torch.float32).view(-1, 1)
        self.hidden_state = hidden


class Network(nn.Module):
    def __init__(self, action_dim: int, obs_shape=(4, 84, 84), hidden_dim: int = 512,
                 encoder: str = "nature", forward_steps: int = 5, mlp_hidden: int = 128):
        super().__init__()
        self.action_dim = action_dim
        self.obs_shape = tuple(obs_shape)
        self.hidden_dim = hidden_dim
        self.max_forward_steps = forward_steps

        self.encoder = make_encoder(encoder, obs_shape, hidden_dim, mlp_hidden)
        self.recurrent = nn.LSTM(self.encoder.out_dim + action_dim + 1, hidden_dim,
                                 batch_first=True)
        self.advantage = nn.Sequential(
            nn.Linear(hidden_dim, hidden_dim), nn.ReLU(True),
            nn.Linear(hidden_dim, action_dim))
        self.value = nn.Sequential(
            nn.Linear(hidden_dim, hidden_dim), nn.ReLU(True),
            nn.Linear(hidden_dim, 1))

    # ------------------------------------------------------------------
    def _dueling_q(self, hidden: torch.Tensor) -> torch.Tensor:
        adv = self.advantage(hidden)
        val = self.value(hidden)
        return val + adv - adv.mean(-1, keepdim=True)

    def _norm_obs(self, obs: torch.Tensor) -> torch.Tensor:
        if obs.dtype == torch.uint8:
            obs = obs.float()
            if len(self.obs_shape) == 3:
                obs = obs / 255.0
        return obs

    # ------------------------------------------------------------------
    def forward(self, state: AgentState):
        """Single-step actor inference.  Returns (q (B, A), (h, c))."""
        obs = self._norm_obs(state.obs)
        latent = self.encoder(obs)
        rin = torch.cat((latent, state.last_action, state.last_reward), dim=1)
        _, hidden = self.recurrent(rin.unsqueeze(1), state.hidden_state)
        q = self._dueling_q(hidden[0].squeeze(0))
        return q, hidden

    # ------------------------------------------------------------------
    def _run_sequence(self, obs, last_action, last_reward, hidden_state, seq_len):
        """Encoder + LSTM over padded (B, T, ...) input; returns (B, T, H)."""
        batch_size, max_seq_len = obs.shape[:2]
        flat_obs = self._norm_obs(obs.reshape(-1, *self.obs_shape))
        latent = self.encoder(flat_obs)
        rin = torch.cat((latent, last_action.reshape(-1, self.action_dim).float(),
                         last_reward.reshape(-1, 1)), dim=1)
        rin = rin.view(batch_size, max_seq_len, -1)
        packed = pack_padded_sequence(rin, seq_len.cpu(), batch_first=True,
                                      enforce_sorted=False)
        out, _ = self.recurrent(packed, hidden_state)
        out, _ = pad_packed_sequence(out, batch_first=True)
        return out

    @staticmethod
    def _gather_positions(out: torch.Tensor, pos: torch.Tensor) -> torch.Tensor:
        """out: (B, T, H); pos: list of per-sample index tensors -> (sum, H)."""
        b_idx = torch.cat([torch.full((len(p),), i, dtype=torch.long)
                           for i, p in enumerate(pos)])
        t_idx = torch.cat(pos)
        return out[b_idx.to(out.device), t_idx.to(out.device)]

    def _target_positions(self, burn_in_steps, learning_steps, forward_steps):
        """Vectorized equivalent of the reference's tail-repeat slice
        (model.py:102-111): per sample, indices
        min(burn_in + n + i, burn_in+learn+fwd-1) for i in [0, learn)."""
        pos = []
        for b, l, f in zip(burn_in_steps.tolist(), learning_steps.tolist(),
                           forward_steps.tolist()):
            idx = torch.arange(l) + b + self.max_forward_steps
            pos.append(torch.clamp(idx, max=b + l + f - 1))
        return pos

    def calculate_q_(self, obs, last_action, last_reward, hidden_state,
                     burn_in_steps, learning_steps, forward_steps):
        """Q at target positions t+n over the full sequence -> (sum_learn, A)."""
        seq_len = burn_in_steps + learning_steps + forward_steps
        out = self._run_sequence(obs, last_action, last_reward, hidden_state, seq_len)
        pos = self._target_positions(burn_in_steps, learning_steps, forward_steps)
        hidden = self._gather_positions(out, pos)
        assert hidden.size(0) == int(learning_steps.sum())
        return self._dueling_q(hidden)

    def calculate_q(self, obs, last_action, last_reward, hidden_state,
                    burn_in_steps, learning_steps):
        """Q at learning positions (burn_in..burn_in+learn) -> (sum_learn, A)."""
        seq_len = burn_in_steps + learning_steps
        out = self._run_sequence(obs, last_action, last_reward, hidden_state, seq_len)
        pos = [torch.arange(l) + b for b, l in
               zip(burn_in_steps.tolist(), learning_steps.tolist())]
        hidden = self._gather_positions(out, pos)
        return self._dueling_q(hidden)

    def calculate_q_both(self, obs, last_action, last_reward, hidden_state,
                         burn_in_steps, learning_steps, forward_steps):
        """ONE full-sequence pass returning (q_learn, q_target_pos), each
        (sum_learn, A).  q_learn equals calculate_q's output and q_target_pos
        equals calculate_q_'s output (same weights, same inputs), computed
        without the duplicate encoder+LSTM pass the reference performs
        (worker.py:346 vs :352)."""
        seq_len = burn_in_steps + learning_steps + forward_steps
        out = self._run_sequence(obs, last_action, last_reward, hidden_state, seq_len)
        pos_learn = [torch.arange(l) + b for b, l in
                     zip(burn_in_steps.tolist(), learning_steps.tolist())]
        pos_tgt = self._target_positions(burn_in_steps, learning_steps, forward_steps)
        h_learn = self._gather_positions(out, pos_learn)
        h_tgt = self._gather_positions(out, pos_tgt)
        return self._dueling_q(h_learn), self._dueling_q(h_tgt)


# ---------------------------------------------------------------------------
# Reference-checkpoint interop
# ---------------------------------------------------------------------------

_REFERENCE_ENCODER_KEYS = {
    # reference model.py:39-49 nn.Sequential indices -> our named modules
    "feature.0": "encoder.conv1",   # Conv2d(1, 32, 8, 4)
    "feature.2": "encoder.conv2",   # Conv2d(32, 64, 4, 2)
    "feature.4": "encoder.conv3",   # Conv2d(64, 64, 3, 1)
    "feature.7": "encoder.fc",      # Linear(3136, 512)
}


def reference_state_dict_to_native(state_dict):
    """Rename a ZiyuanMa/R2D2 reference ``Network`` state_dict to this
    framework's module names.  Only the encoder differs (the reference's
    anonymous ``feature`` Sequential, model.py:39-49); ``recurrent``,
    ``advantage`` and ``value`` already share names and structure.  The
    result loads into ``Network(action_dim, obs_shape=(1, 84, 84))`` —
    the reference's single-grayscale-frame architecture."""
    out = {}
    for k, v in state_dict.items():
        head, _, tail = k.rpartition(".")   # e.g. "feature.0", "weight"
        out[f"{_REFERENCE_ENCODER_KEYS.get(head, head)}.{tail}"
            if head else k] = v
    return out


def load_reference_checkpoint(path, map_location="cpu"):
    """Load a reference 4-tuple checkpoint (test.py:27 contract) into a
    native ``Network``.  Returns (network, num_updates, env_steps, minutes)."""
    state, num_updates, env_steps, minutes = torch.load(
        path, map_location=map_location, weights_only=False)
    state = reference_state_dict_to_native(state)
    action_dim = state["advantage.2.weight"].shape[0]
    hidden_dim = state["value.0.weight"].shape[0]
    net = Network(action_dim, obs_shape=(1, 84, 84), hidden_dim=hidden_dim,
                  encoder="nature")
    net.load_state_dict(state)
    return net, int(num_updates), int(env_steps), float(minutes)
