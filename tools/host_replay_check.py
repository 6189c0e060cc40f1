"""configs[1]-style check on one GPU: the REFERENCE process layout (CPU
actor processes + host ReplayBuffer process + GPU learner over mp.Queues)
must train end-to-end — the topology the host-replay presets select."""

import sys

sys.path.insert(0, ".")


def main(updates=60):
    from r2d2_amd import config as cfg

    cfg.apply("mspacman", gpu_replay=False, num_actors=4,
              buffer_capacity=40_000, learning_starts=4_000,
              training_steps=updates, log_interval=5,
              save_interval=100_000, batch_queue_size=4)
    from r2d2_amd.train import train
    train(seed=0)
    print("host-replay topology OK")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 60)
