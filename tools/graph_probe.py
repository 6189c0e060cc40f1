"""Bisect hipGraph compatibility of the engine's kernels: capture pieces
and replay, checking outputs + the LSTM poison word."""
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
    X0 = torch.randn(B, T, 4*H, device=dev).bfloat16()
    X1 = torch.randn(B, T, 4*H, device=dev).bfloat16()
    W0 = (torch.randn(4*H, H, device=dev)*0.02).bfloat16()
    W1 = (torch.randn(4*H, H, device=dev)*0.02).bfloat16()
    init = torch.zeros(2, B, H, device=dev)
    lens = torch.full((B,), T, dtype=torch.int32, device=dev)
    bar = torch.zeros(512, dtype=torch.int32, device=dev)

    def fwd():
        return m.lstm_fwd(X0, X1, W0, W1, init, init, lens, bar, True)

    # eager reference
    out = fwd(); torch.cuda.synchronize()
    ref = out[0].clone()
    print("eager ok, poison:", int(bar[256].item()))

    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        fwd()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        outs = fwd()
    print("captured")
    for i in range(5):
        g.replay()
        torch.cuda.synchronize()
        print(f"replay {i}: poison={int(bar[256].item())}, "
              f"match={torch.equal(outs[0], ref)}")

if __name__ == "__main__":
    main()
