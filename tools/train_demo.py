"""Live training demo on one MI355X through the REAL train() orchestration
(configs[2]): a spawned VectorActor process (K15 HIP inference) feeds the
learner process's GPU-resident replay over mp.Queues; the learner trains
through the HIP engine.  JSONL metrics land in gpurun_out/."""

import sys

sys.path.insert(0, ".")


def main(updates=500, actors=64, preset="mspacman_gpu_replay"):
    from r2d2_amd import config as cfg

    overrides = {}
    if preset == "seaquest_impala":   # configs[4] live topology
        overrides = dict(gpu_replay=True, vector_actors=True,
                         actor_device="cuda")
    cfg.apply(preset, **overrides, num_actors=actors,
              buffer_capacity=400_000, learning_starts=20_000,
              training_steps=updates, log_interval=5,
              save_interval=100_000, actor_update_interval=400,
              metrics_path="gpurun_out/train_demo_metrics.jsonl")
    from r2d2_amd.train import train
    train(seed=0)
    print("demo done")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 500,
         int(sys.argv[2]) if len(sys.argv) > 2 else 64,
         sys.argv[3] if len(sys.argv) > 3 else "mspacman_gpu_replay")
