"""Real-reward learning validation on the native CartPole env (configs[0]):
trains the full R2D2 stack (prioritized replay, burn-in LSTM, double-Q,
n-step) on CPU and emits the reward curve — the reference's only published
evidence is a learning curve (images/MsPacman.jpg); this is the same
artifact on the env this image can actually run end-to-end."""

import json
import sys

sys.path.insert(0, ".")


def main(training_steps=8000):
    from r2d2_amd import config as cfg

    cfg.apply("cartpole", training_steps=training_steps,
              learning_starts=1500, buffer_capacity=40_000,
              max_episode_steps=500, log_interval=10,
              save_interval=100_000, num_actors=2,
              base_eps=0.15, lr=1e-3,
              metrics_path="gpurun_out/cartpole_metrics.jsonl")
    from r2d2_amd.train import train
    train(seed=0)
    rows = [json.loads(l) for l in
            open("gpurun_out/cartpole_metrics.jsonl")]
    rets = [r["avg_episode_return"] for r in rows
            if "avg_episode_return" in r]
    print("reward curve:", [round(r, 1) for r in rets])


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 8000)
