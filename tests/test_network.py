import numpy as np
import pytest
import torch

from r2d2_amd.models.network import Network, AgentState


def make_batch(net, B=3, burn=(0, 2, 4), learn=(4, 4, 3), fwd=(2, 2, 1),
               obs_shape=(1, 84, 84), seed=0):
    torch.manual_seed(seed)
    burn = torch.tensor(burn)
    learn = torch.tensor(learn)
    fwd = torch.tensor(fwd)
    T = int((burn + learn + fwd).max())
    if len(obs_shape) == 3:
        obs = torch.randint(0, 255, (B, T) + obs_shape, dtype=torch.uint8)
    else:
        obs = torch.randn(B, T, obs_shape[0])
    A = net.action_dim
    la = torch.zeros(B, T, A)
    la[:, :, 0] = 1
    lr = torch.randn(B, T) * 0.1
    h0 = (torch.zeros(1, B, net.hidden_dim), torch.zeros(1, B, net.hidden_dim))
    return obs, la, lr, h0, burn, learn, fwd


@pytest.fixture(scope="module")
def small_net():
    torch.manual_seed(0)
    return Network(action_dim=5, obs_shape=(1, 84, 84), hidden_dim=32,
                   encoder="nature", forward_steps=2)


def test_calculate_q_shapes(small_net):
    obs, la, lr, h0, burn, learn, fwd = make_batch(small_net)
    q = small_net.calculate_q(obs, la, lr, h0, burn, learn)
    assert q.shape == (int(learn.sum()), 5)
    q_ = small_net.calculate_q_(obs, la, lr, h0, burn, learn, fwd)
    assert q_.shape == (int(learn.sum()), 5)


def test_calculate_q_both_matches_separate(small_net):
    obs, la, lr, h0, burn, learn, fwd = make_batch(small_net)
    with torch.no_grad():
        q_learn, q_tgt = small_net.calculate_q_both(obs, la, lr, h0, burn, learn, fwd)
        q_sep = small_net.calculate_q(obs, la, lr, h0, burn, learn)
        q_sep_ = small_net.calculate_q_(obs, la, lr, h0, burn, learn, fwd)
    assert torch.allclose(q_learn, q_sep, atol=1e-5)
    assert torch.allclose(q_tgt, q_sep_, atol=1e-5)


def test_target_positions_tail_repeat(small_net):
    """forward_steps < max_forward_steps repeats the last hidden (the
    reference's tail-padding, model.py:102-111)."""
    # one sample, learn=4, fwd=1 < max_forward 2 => last position repeated
    obs, la, lr, h0, burn, learn, fwd = make_batch(
        small_net, B=1, burn=(0,), learn=(4,), fwd=(1,))
    with torch.no_grad():
        q_ = small_net.calculate_q_(obs, la, lr, h0, burn, learn, fwd)
    assert q_.shape == (4, 5)
    # positions: min(0+2+i, 4) for i in 0..3 -> 2,3,4,4 (last repeated)
    assert torch.allclose(q_[2], q_[3], atol=1e-6)
    assert not torch.allclose(q_[0], q_[1], atol=1e-4)


def test_forward_single_step_matches_sequence(small_net):
    """Stepping one-by-one through forward() must equal the batched sequence
    run (same LSTM trajectory)."""
    T, A, H = 4, 5, 32
    torch.manual_seed(1)
    obs = torch.randint(0, 255, (1, T, 1, 84, 84), dtype=torch.uint8)
    actions = [0, 2, 1, 3]
    rewards = [0.0, 1.0, -0.5, 0.25]
    # single-step path
    state = AgentState(obs[0, :1], A)
    qs = []
    with torch.no_grad():
        for t in range(T):
            q, hidden = small_net(state)
            qs.append(q)
            if t + 1 < T:
                state.update(obs[0, t + 1: t + 2], actions[t], [rewards[t]], hidden)
    # sequence path: same last_action/last_reward stream
    la = torch.zeros(1, T, A)
    la[0, 0, 0] = 1
    for t in range(1, T):
        la[0, t, actions[t - 1]] = 1
    lr = torch.tensor([[0.0] + rewards[:-1]])
    h0 = (torch.zeros(1, 1, H), torch.zeros(1, 1, H))
    with torch.no_grad():
        q_seq = small_net.calculate_q(obs, la, lr, h0,
                                      torch.tensor([0]), torch.tensor([T]))
    stepped = torch.cat(qs)
    assert torch.allclose(stepped, q_seq, atol=1e-4)


def test_mlp_and_impala_encoders():
    torch.manual_seed(0)
    mlp_net = Network(2, (4,), hidden_dim=16, encoder="mlp", forward_steps=2,
                      mlp_hidden=16)
    obs, la, lr, h0, burn, learn, fwd = make_batch(mlp_net, obs_shape=(4,))
    h0 = (torch.zeros(1, 3, 16), torch.zeros(1, 3, 16))
    q = mlp_net.calculate_q(obs, la, lr, h0, burn, learn)
    assert q.shape == (int(learn.sum()), 2)

    imp_net = Network(4, (4, 84, 84), hidden_dim=32, encoder="impala",
                      forward_steps=2)
    obs, la, lr, h0, burn, learn, fwd = make_batch(
        imp_net, B=2, burn=(0, 1), learn=(2, 2), fwd=(1, 1), obs_shape=(4, 84, 84))
    q = imp_net.calculate_q_(obs, la, lr, h0, burn, learn, fwd)
    assert q.shape == (4, 4)


def test_agent_state_no_shared_default():
    """The reference's AgentState shares one class-level last_reward tensor
    across instances (model.py:14); ours must not."""
    s1 = AgentState(torch.zeros(1, 4), 2)
    s2 = AgentState(torch.zeros(1, 4), 2)
    s1.last_reward[0, 0] = 99.0
    assert s2.last_reward[0, 0] == 0.0


def test_reference_checkpoint_key_mapping_roundtrip(tmp_path):
    """A reference-named state_dict (anonymous `feature` Sequential,
    reference model.py:39-49) must load into our Network bit-exactly via
    reference_state_dict_to_native / load_reference_checkpoint."""
    import torch

    from r2d2_amd.models.network import (Network, load_reference_checkpoint,
                                         reference_state_dict_to_native)

    torch.manual_seed(11)
    src = Network(9, obs_shape=(1, 84, 84), hidden_dim=512, encoder="nature")
    inv = {"encoder.conv1": "feature.0", "encoder.conv2": "feature.2",
           "encoder.conv3": "feature.4", "encoder.fc": "feature.7"}
    ref_sd = {}
    for k, v in src.state_dict().items():
        head, _, tail = k.rpartition(".")
        ref_sd[f"{inv.get(head, head)}.{tail}" if head else k] = v
    assert any(k.startswith("feature.") for k in ref_sd)

    native = reference_state_dict_to_native(ref_sd)
    dst = Network(9, obs_shape=(1, 84, 84), hidden_dim=512, encoder="nature")
    dst.load_state_dict(native)
    for k, v in src.state_dict().items():
        assert torch.equal(v, dst.state_dict()[k]), k

    # full 4-tuple loader (reference test.py:27 contract)
    p = tmp_path / "MsPacman500.pth"
    torch.save((ref_sd, 500, 123456, 7.5), p)
    net, nu, es, mins = load_reference_checkpoint(str(p))
    assert (nu, es, mins) == (500, 123456, 7.5)
    assert torch.equal(net.state_dict()["encoder.conv1.weight"],
                       src.state_dict()["encoder.conv1.weight"])
