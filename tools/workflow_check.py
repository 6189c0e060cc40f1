"""End-to-end reference-workflow check on one GPU: train() (configs[2]
topology) -> 4-tuple checkpoints -> evaluate.test() learning-curve emit.
Must be a real file (not stdin): the spawn-context actor child re-imports
__main__."""

import sys

sys.path.insert(0, ".")


def main():
    from r2d2_amd import config as cfg
    cfg.apply("mspacman_gpu_replay", num_actors=32, buffer_capacity=200_000,
              learning_starts=10_000, training_steps=120, log_interval=5,
              save_interval=40, actor_update_interval=400)
    from r2d2_amd.train import train
    train(seed=0)
    print("train done; evaluating checkpoints")
    cfg.apply("mspacman_gpu_replay", save_interval=40, max_episode_steps=400)
    from r2d2_amd.evaluate import test
    rows = test(model_dir="models", num_episodes=3, out_dir="gpurun_out")
    print("eval rows:", len(rows))
    assert len(rows) >= 2
    print("workflow OK")


if __name__ == "__main__":
    main()
