"""Bisect hipGraph compatibility of the persistent LSTM: check whether the
captured hipMemsetAsync (counter reset) actually replays, by CHANGING the
input between replays (stale counters would skip the handoff waits and
produce wrong outputs) and by dumping the counter words."""
import sys
sys.path.insert(0, ".")
import torch
from r2d2_amd import config as cfg

def main():
    c = cfg.apply("mspacman")
    from r2d2_amd.ops import hip_ops
    m = hip_ops.ext()
    dev = torch.device("cuda")
    B, T, H = 64, 85, 512
    g_ = torch.Generator(device=dev).manual_seed(0)
    def rnd():
        return torch.randn(B, T, 4*H, device=dev, generator=g_).bfloat16()
    X0 = rnd(); X1 = rnd()
    W0 = (torch.randn(4*H, H, device=dev, generator=g_)*0.02).bfloat16()
    W1 = W0.clone()
    init = torch.zeros(2, B, H, device=dev)
    lens = torch.full((B,), T, dtype=torch.int32, device=dev)
    bar = torch.zeros(512, dtype=torch.int32, device=dev)

    def fwd():
        return m.lstm_fwd(X0, X1, W0, W1, init, init, lens, bar, True)

    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        fwd()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        outs = fwd()
    print("captured")

    for i in range(4):
        xa = rnd()
        X0.copy_(xa)
        g.replay(); torch.cuda.synchronize()
        got = outs[0].clone()
        # eager reference with the same input on fresh counters
        bar2 = torch.zeros(512, dtype=torch.int32, device=dev)
        ref = m.lstm_fwd(X0, X1, W0, W1, init, init, lens, bar2, True)[0]
        torch.cuda.synchronize()
        ctrs = [int(bar[j].item()) for j in (0, 32, 64, 96, 256)]
        print(f"replay {i}: ctrs={ctrs} "
              f"match={torch.equal(got, ref)} "
              f"maxdiff={float((got.float()-ref.float()).abs().max())}")

if __name__ == "__main__":
    main()
