"""Short live training demo on one MI355X: VectorActor (K15 HIP inference)
feeding the learner-owned GPU-resident replay, HIP engine updates, JSONL
metrics.  Evidence run for the configs[2] topology (see profiles/README)."""

import sys
import threading

sys.path.insert(0, ".")

import torch  # noqa: E402

from r2d2_amd import config as cfg  # noqa: E402
from r2d2_amd.models.network import Network  # noqa: E402
from r2d2_amd.train import epsilon_ladder  # noqa: E402
from r2d2_amd.worker import Learner, VectorActor  # noqa: E402

import queue  # noqa: E402


def main(updates=600, actors=64):
    c = cfg.apply("mspacman_gpu_replay", num_actors=actors,
                  buffer_capacity=400_000, learning_starts=20_000,
                  training_steps=updates, log_interval=5,
                  save_interval=100_000, actor_update_interval=400,
                  metrics_path="gpurun_out/train_demo_metrics.jsonl")
    torch.manual_seed(0)
    model = Network(c.action_dim, c.obs_shape, c.hidden_dim,
                    encoder=c.encoder, forward_steps=c.forward_steps)
    model.share_memory()
    sq = queue.Queue()
    learner = Learner(None, None, model, model_dir="gpurun_out/demo_models")
    learner.enable_hip_engine()
    assert learner.engine is not None

    va = VectorActor(epsilon_ladder(c.num_actors), model, [sq],
                     device="cuda", seed=1)
    assert va.hip_inf is not None
    stop = threading.Event()

    def drive():
        while not stop.is_set():
            va.run(stop_after_steps=500)

    t = threading.Thread(target=drive)
    t.start()
    try:
        learner.run_with_gpu_replay([sq])
    finally:
        stop.set()
        t.join(timeout=60)
        torch.cuda.synchronize()
    print(f"demo done: {learner.num_updates} updates, "
          f"{learner.env_steps} env steps")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 600,
         int(sys.argv[2]) if len(sys.argv) > 2 else 64)
