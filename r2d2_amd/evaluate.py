"""Policy-evaluation harness (reference: /root/reference/test.py:14-88).

Iterates saved checkpoints ``models/{game}{k*save_interval}.pth`` (the
reference 4-tuple format: (state_dict, num_updates, env_steps, minutes) —
reference worker.py:380-381 / test.py:27), replays each through
``num_episodes`` parallel near-greedy episodes (eps = config.eval_eps,
reference test.py:18,32-33 runs 5 episodes in mp.Pool(5)), and emits the
learning curve — reward vs env frames (= env_steps * frame_skip,
reference test.py:28,36) and vs wall-clock hours (test.py:29).

Curves are written as ``{game}_eval.csv`` and ``{game}_eval.jsonl``, plus
the reference's ``{game}.jpg`` two-panel plot (reward vs env frames and vs
wall clock, reference test.py:42-58) when matplotlib is importable.
"""

import csv
import json
import os
from typing import Optional

import numpy as np
import torch
import torch.multiprocessing as mp

from . import config as cfg
from .models.network import AgentState, Network


def test_one_case(args):
    """One near-greedy episode under a checkpointed policy.

    ``args`` = (state_dict, seed, config_dict).  Returns
    (episode_reward, episode_steps).  Reference: test.py:60-81.
    The config snapshot rides along because pool workers are spawned
    (fresh interpreters) and would otherwise see default config state.
    """
    state_dict, seed, config_dict = args
    c = cfg.apply(**config_dict) if config_dict else cfg.get()
    torch.set_num_threads(1)
    from .envs import create_env

    env = create_env(seed=seed)
    network = Network(env.action_dim, c.obs_shape, c.hidden_dim,
                      encoder=c.encoder, forward_steps=c.forward_steps,
                      mlp_hidden=c.mlp_hidden)
    try:
        network.load_state_dict(state_dict)
    except RuntimeError:
        # checkpoint trained by the ORIGINAL reference: anonymous `feature`
        # Sequential encoder keys (reference model.py:39-49) — remap
        from .models.network import reference_state_dict_to_native
        network.load_state_dict(reference_state_dict_to_native(state_dict))
    network.eval()
    rng = np.random.default_rng(seed)

    obs = env.reset()
    state = AgentState(torch.from_numpy(obs).unsqueeze(0), env.action_dim)
    done = False
    episode_reward, episode_steps = 0.0, 0
    while not done and episode_steps < c.max_episode_steps:
        with torch.no_grad():
            q, hidden = network(state)
        if rng.random() < c.eval_eps:
            action = int(rng.integers(env.action_dim))
        else:
            action = int(torch.argmax(q, 1).item())
        obs, reward, done, _ = env.step(action)
        state.update(obs[None], action, [reward], hidden)
        episode_reward += reward
        episode_steps += 1
    return episode_reward, episode_steps


def _checkpoint_paths(model_dir: str, game_name: str, save_interval: int):
    """Yield (num_updates, path) in update order.

    The reference (test.py:18-20) steps k=500,1000,... and stops at the
    first missing file — which silently skips EVERYTHING when the run
    used a different save_interval, or stops early at any gap.  Glob the
    actual ``{game}{N}.pth`` files and sort numerically instead (the
    ``.train.pth`` optimizer sidecars are not checkpoints)."""
    import glob
    import re

    found = []
    for path in glob.glob(os.path.join(model_dir, f"{game_name}*.pth")):
        if path.endswith(".train.pth"):
            continue
        m = re.fullmatch(re.escape(game_name) + r"(\d+)\.pth",
                         os.path.basename(path))
        if m:
            found.append((int(m.group(1)), path))
    yield from sorted(found)


def test(game_name: Optional[str] = None, model_dir: str = "models",
         num_episodes: Optional[int] = None, pool_size: Optional[int] = None,
         out_dir: Optional[str] = None):
    """Evaluate every checkpoint and write the learning curve.

    Returns the list of per-checkpoint result dicts.  Reference: test.py:14-58.
    """
    c = cfg.get()
    game_name = game_name or c.game_name
    num_episodes = num_episodes or 5                 # reference test.py:18
    pool_size = pool_size or num_episodes
    out_dir = out_dir or model_dir

    results = []
    ckpts = list(_checkpoint_paths(model_dir, game_name, c.save_interval))
    if not ckpts:
        print(f"no checkpoints found under {model_dir}/{game_name}*.pth")
        return results

    from dataclasses import asdict
    config_dict = asdict(c)
    ctx = mp.get_context("spawn")
    with ctx.Pool(pool_size) as pool:
        for num_updates, path in ckpts:
            state_dict, saved_updates, env_steps, minutes = torch.load(
                path, map_location="cpu", weights_only=False)
            args = [(state_dict, 10_000 + num_updates + i, config_dict)
                    for i in range(num_episodes)]
            rewards_steps = pool.map(test_one_case, args)
            rewards = [r for r, _ in rewards_steps]
            row = {
                "num_updates": int(saved_updates),
                "env_steps": int(env_steps),
                "env_frames": int(env_steps) * c.frame_skip,  # test.py:28,36
                "training_hours": float(minutes) / 60.0,      # test.py:29
                "mean_reward": float(np.mean(rewards)),
                "std_reward": float(np.std(rewards)),
                "rewards": [float(r) for r in rewards],
            }
            results.append(row)
            print(f"{game_name}{num_updates}: mean reward "
                  f"{row['mean_reward']:.1f} +/- {row['std_reward']:.1f} "
                  f"({row['env_frames']} frames, {row['training_hours']:.2f} h)")

    os.makedirs(out_dir, exist_ok=True)
    csv_path = os.path.join(out_dir, f"{game_name}_eval.csv")
    with open(csv_path, "w", newline="") as f:
        w = csv.DictWriter(f, fieldnames=["num_updates", "env_steps",
                                          "env_frames", "training_hours",
                                          "mean_reward", "std_reward"])
        w.writeheader()
        for row in results:
            w.writerow({k: row[k] for k in w.fieldnames})
    with open(os.path.join(out_dir, f"{game_name}_eval.jsonl"), "w") as f:
        for row in results:
            f.write(json.dumps(row) + "\n")

    _maybe_plot(results, game_name, out_dir)
    return results


def _maybe_plot(results, game_name, out_dir):
    """Reward vs env-frames and vs wall-clock plot (reference test.py:42-58);
    silently skipped when matplotlib is absent."""
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
    except ImportError:
        return
    fig, (ax1, ax2) = plt.subplots(1, 2, figsize=(12, 4.5))
    frames = [r["env_frames"] for r in results]
    hours = [r["training_hours"] for r in results]
    rewards = [r["mean_reward"] for r in results]
    ax1.plot(frames, rewards)
    ax1.set_xlabel("environment frames")
    ax1.set_ylabel("average reward")
    ax2.plot(hours, rewards)
    ax2.set_xlabel("training time (hours)")
    fig.suptitle(game_name)
    fig.savefig(os.path.join(out_dir, f"{game_name}.jpg"))
    plt.close(fig)


def main():
    import argparse
    ap = argparse.ArgumentParser(description="Evaluate saved checkpoints "
                                 "(reference test.py harness)")
    ap.add_argument("--game", type=str, default=None)
    ap.add_argument("--preset", type=str, default=None)
    ap.add_argument("--model-dir", type=str, default="models")
    ap.add_argument("--episodes", type=int, default=5)
    ap.add_argument("--out-dir", type=str, default=None)
    args = ap.parse_args()
    if args.preset:
        cfg.apply(args.preset)
    test(game_name=args.game, model_dir=args.model_dir,
         num_episodes=args.episodes, out_dir=args.out_dir)


if __name__ == "__main__":
    main()
