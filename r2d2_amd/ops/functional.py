"""Pure-functional numerics of the R2D2 algorithm.

Torch versions are the golden references the HIP kernels are unit-tested
against; numpy versions serve the (CPU) actor/LocalBuffer path.

Reference semantics reproduced here:
- value_rescale / inverse_value_rescale: h(x) = sign(x)(sqrt(|x|+1)-1) + eps*x
  and its closed-form inverse (reference: worker.py:383-390).
- n-step discounted return via convolution with a gamma kernel
  (reference: worker.py:466-469).
- per-step gamma^n vector with bootstrap-cut decay and 0 at terminal
  (reference: worker.py:443-455).
- mixed TD priority  eta*max + (1-eta)*mean over ragged segments
  (reference: worker.py:268-276).
"""

import numpy as np
import torch


# ---------------------------------------------------------------------------
# Value rescaling (R2D2 paper eq. h / h^-1)
# ---------------------------------------------------------------------------

def value_rescale(x: torch.Tensor, eps: float = 1e-3) -> torch.Tensor:
    return x.sign() * ((x.abs() + 1).sqrt() - 1) + eps * x


def inverse_value_rescale(x: torch.Tensor, eps: float = 1e-3) -> torch.Tensor:
    t = (1 + 4 * eps * (x.abs() + 1 + eps)).sqrt() - 1
    return x.sign() * ((t / (2 * eps)).square() - 1)


# ---------------------------------------------------------------------------
# Loss
# ---------------------------------------------------------------------------

def per_step_loss(q: torch.Tensor, target: torch.Tensor, kind: str = "huber",
                  kappa: float = 1.0) -> torch.Tensor:
    """Elementwise TD loss (no reduction). 'mse' is the reference-compat mode
    (reference: worker.py:290 uses MSELoss(reduction='none'))."""
    td = q - target
    if kind == "mse":
        return td.square()
    a = td.abs()
    return torch.where(a <= kappa, 0.5 * td.square(), kappa * (a - 0.5 * kappa))


# ---------------------------------------------------------------------------
# n-step machinery (numpy; runs on actors inside LocalBuffer.finish())
# ---------------------------------------------------------------------------

def n_step_return(rewards: np.ndarray, n: int, gamma: float) -> np.ndarray:
    """rewards: (T,) float32 of raw per-step rewards.  Returns (T,) where
    out[t] = sum_{i=0}^{min(n, T-t)-1} gamma^i * rewards[t+i].

    Matches the reference's np.convolve construction (worker.py:466-469):
    trailing steps use fewer than n terms (episode/bootstrap cut).
    """
    T = len(rewards)
    kernel = gamma ** np.arange(n, dtype=np.float64)
    # full convolution of reversed kernel; take the aligned window
    out = np.convolve(rewards.astype(np.float64), kernel[::-1], mode="full")[n - 1: n - 1 + T]
    return out.astype(np.float32)


def gamma_vector(T: int, n: int, gamma: float, done: bool) -> np.ndarray:
    """Per-step bootstrap discount gamma^n_t (reference: worker.py:443-455).

    For step t the bootstrap target is Q(s_{t+n}) discounted by gamma^n.
    - mid-episode steps (t + n <= T-?): gamma^n
    - at an episode cut without terminal (bootstrap cut at T): the last n-1
      steps bootstrap from the final state with decaying exponents
      gamma^{T-t} ... (fewer real rewards were accumulated)
    - if the episode ended with a terminal at step T: the last n steps have
      NO bootstrap -> gamma^n_t = 0 (this is how the reference avoids storing
      a `done` flag at all).
    """
    out = np.full(T, gamma ** n, dtype=np.float32)
    if done:
        out[max(0, T - n):] = 0.0
    else:
        for t in range(max(0, T - n + 1), T):
            out[t] = gamma ** (T - t)
    return out


# ---------------------------------------------------------------------------
# Mixed max/mean TD priority (ragged segments)
# ---------------------------------------------------------------------------

def mixed_td_priority_np(abs_td: np.ndarray, lengths: np.ndarray,
                         eta: float = 0.9) -> np.ndarray:
    """abs_td: flat (sum(lengths),) of |TD|; lengths: (B,) segment lengths.
    Returns (B,) priorities = eta*max + (1-eta)*mean per segment
    (reference: worker.py:268-276, the numpy loop)."""
    out = np.empty(len(lengths), dtype=np.float32)
    ofs = 0
    # int() per element: uint8 lengths (the Block storage dtype) would
    # otherwise wrap the running offset at 256 for full 400-step blocks
    for i, L in enumerate(lengths):
        L = int(L)
        seg = abs_td[ofs: ofs + L]
        out[i] = eta * seg.max() + (1.0 - eta) * seg.mean()
        ofs += L
    return out


def mixed_td_priority(abs_td: torch.Tensor, lengths: torch.Tensor,
                      eta: float = 0.9) -> torch.Tensor:
    """Torch golden version (segmented reduce)."""
    outs = []
    ofs = 0
    for L in lengths.tolist():
        seg = abs_td[ofs: ofs + L]
        outs.append(eta * seg.max() + (1.0 - eta) * seg.mean())
        ofs += L
    return torch.stack(outs)


# ---------------------------------------------------------------------------
# Double-Q target assembly (golden; the HIP fused kernel replicates this)
# ---------------------------------------------------------------------------

def double_q_target(q_online_tgt_pos: torch.Tensor, q_target_tgt_pos: torch.Tensor,
                    n_step_reward: torch.Tensor, gamma_n: torch.Tensor,
                    eps: float = 1e-3) -> torch.Tensor:
    """target = h( r_n + gamma^n * h^-1( Q_target[argmax_a Q_online] ) )
    (reference: worker.py:346-349). Inputs are at target positions (t+n)."""
    a_star = q_online_tgt_pos.argmax(dim=-1, keepdim=True)
    q_next = q_target_tgt_pos.gather(-1, a_star).squeeze(-1)
    return value_rescale(n_step_reward + gamma_n * inverse_value_rescale(q_next, eps), eps)
