"""IMPALA-deep ResNet encoder on the gfx950 HIP kernels (configs[4]).

Runs the ImpalaCNN (models/encoders.py — conv 3x3 + maxpool 3x3 s2 + 2
pre-activation residual blocks per stage, channels (16, 32, 32), then
relu -> flatten -> fc 3872->512) through ops/hip/impala_kernels.hip:
halo-padded NHWC activations, MFMA implicit-GEMM 3x3 convs whose dgrad is
the same kernel with flipped prepacked weights, fused pre-activation
ReLU / residual epilogues, and an atomic-free maxpool backward.

The engine (ops/engine.py) delegates encoder fwd/bwd here when
config.encoder == 'impala'; LSTM / heads / loss / optimizer paths are
shared with the flagship nature config.
"""

from typing import Optional

import torch

# stage geometry: (conv-input H, pooled H) per stage; 84 -> 42 -> 21 -> 11
STAGES = ((84, 42), (42, 21), (21, 11))
CHANS = (16, 32, 32)


class _Arena:
    """Step-scoped reusable device buffers.  Padded conv outputs need a
    zero halo that the kernels never touch — so each buffer is allocated
    (and zeroed) ONCE and reused every step, eliminating the per-step
    FillFunctor storm (~2 ms/step at M=5440).  Buffers are handed out in
    call order within a tag; reset(tag) rewinds that tag's counters."""

    def __init__(self, device):
        self.device = device
        self.bufs = {}
        self.idx = {}

    def reset(self, tag):
        for k in self.idx:
            if k[0] == tag:
                self.idx[k] = 0

    def get(self, tag, shape, dtype=torch.bfloat16):
        key = (tag,) + tuple(shape) + (dtype,)
        lst = self.bufs.setdefault(key, [])
        i = self.idx.get(key, 0)
        if i >= len(lst):
            lst.append(torch.zeros(*shape, device=self.device, dtype=dtype))
        self.idx[key] = i + 1
        return lst[i]


def _pack_fwd(w: torch.Tensor, cp: int, dev) -> torch.Tensor:
    """(COUT, CIN, 3, 3) -> (COUT, 9*cp) bf16, k-order (ky, kx, c),
    input channels zero-padded to cp."""
    cout, cin = w.shape[:2]
    wt = torch.zeros(cout, 3, 3, cp, device=dev)
    wt[:, :, :, :cin] = w.detach().to(dev).permute(0, 2, 3, 1)
    return wt.reshape(cout, 9 * cp).bfloat16().contiguous()


def _pack_dgrad(w: torch.Tensor, dev) -> torch.Tensor:
    """(COUT, CIN, 3, 3) -> (CIN, 9*COUT) bf16: spatially flipped,
    transposed — dgrad of a 3x3 s1 p1 conv as a forward conv over the
    halo-padded upstream gradient."""
    wd = w.detach().to(dev).flip(2, 3).permute(1, 2, 3, 0)
    return wd.reshape(w.shape[1], -1).bfloat16().contiguous()


class ImpalaPack:
    """Prepacked weights for one ImpalaCNN (15 convs + biases)."""

    def __init__(self, enc, device, with_bwd: bool):
        self.enc = enc
        self.device = device
        self.with_bwd = with_bwd
        self.arena = _Arena(device)
        self.refresh()

    def _convs(self):
        """Yield (name, conv, cin_pad) over the 15 convs in forward order:
        per stage: stage conv, res1.conv1, res1.conv2, res2.conv1, res2.conv2."""
        for si, stage in enumerate(self.enc.stages):
            cp = max(8, stage.conv.weight.shape[1])
            yield f"s{si}c", stage.conv, cp
            for ri, res in enumerate((stage.res1, stage.res2)):
                c = res.conv1.weight.shape[1]
                yield f"s{si}r{ri}a", res.conv1, c
                yield f"s{si}r{ri}b", res.conv2, c

    def refresh(self):
        dev = self.device
        self.wt = {}
        self.bias = {}
        self.wd = {}
        for name, conv, cp in self._convs():
            self.wt[name] = _pack_fwd(conv.weight, cp, dev)
            self.bias[name] = conv.bias.detach().to(dev).float().contiguous()
            if self.with_bwd:
                self.wd[name] = _pack_dgrad(conv.weight, dev)
        # fc: torch flattens NCHW (32, 11, 11); our pad2dense flattens HWC
        fc = self.enc.fc
        hd = fc.weight.shape[0]
        wf = fc.weight.detach().to(dev).reshape(hd, 32, 11, 11)
        self.wft = (wf.permute(0, 2, 3, 1).reshape(hd, 32 * 121)
                    .bfloat16().contiguous())
        self.bf = fc.bias.detach().to(dev).float().contiguous()
        if self.with_bwd:
            self.wf_kn = self.wft.t().contiguous()


def pack_obs(m, pack: ImpalaPack, obs_hwc_u8: torch.Tensor) -> torch.Tensor:
    """Frames -> halo-padded 8-channel u8 via the pack's arena; call once
    per step and share between the online and target forward passes."""
    M = obs_hwc_u8.shape[0]
    pack.arena.reset("xp")
    xp = pack.arena.get("xp", (M, 86, 86, 8), torch.uint8)
    m.pack_frames(obs_hwc_u8, xp, 84, 84)
    return xp


def encoder_fwd(m, pack: ImpalaPack, obs_hwc_u8: torch.Tensor,
                want_stash: bool, xp: Optional[torch.Tensor] = None):
    """obs: (M, 84, 84, C<=8) u8 dense -> latent (M, hidden) bf16 (+stash).

    Stash layout (one dict): padded activations and pool argmaxes needed by
    encoder_bwd.  Pass a prepacked ``xp`` (from pack_obs) to skip the frame
    packing — the engine packs once and shares it with the target net."""
    ar = pack.arena
    M = obs_hwc_u8.shape[0] if xp is None else xp.shape[0]
    ar.reset("fwd")
    if xp is None:
        xp = pack_obs(m, pack, obs_hwc_u8)
    st = {"xp": xp} if want_stash else None
    x = xp
    empty = torch.Tensor()
    import os
    fused_pool = os.environ.get("R2D2_IMPALA_FUSED_POOL", "1") != "0"
    for si, (hin, hout) in enumerate(STAGES):
        c = CHANS[si]
        pooled = ar.get("fwd", (M, hout + 2, hout + 2, c))
        arg = ar.get("fwd", (M, hout, hout, c), torch.uint8)
        if fused_pool:
            # stage conv + maxpool in one kernel: the (hin+2)^2 conv output
            # stays in LDS (s0 alone: ~2.6 GB/step of HBM traffic saved
            # across both nets; docs/IMPALA_ROOFLINE.md)
            m.conv3p_pool(x, pack.wt[f"s{si}c"], pack.bias[f"s{si}c"],
                          pooled, arg, M, si)
        else:
            conv_out = ar.get("fwd", (M, hin + 2, hin + 2, c))
            m.conv3p(x, pack.wt[f"s{si}c"], pack.bias[f"s{si}c"], empty,
                     empty, conv_out, M, hin, hin, False, True, 0)
            m.maxpool3s2_fwd(conv_out, pooled, arg, M, hin, hin)
        if want_stash:
            st[f"s{si}c_in"] = x
            st[f"s{si}arg"] = arg
        x = pooled
        for ri in range(2):
            y1 = ar.get("fwd", (M, hout + 2, hout + 2, c))
            m.conv3p(x, pack.wt[f"s{si}r{ri}a"], pack.bias[f"s{si}r{ri}a"],
                     empty, empty, y1, M, hout, hout, True, True, 0)
            out = ar.get("fwd", (M, hout + 2, hout + 2, c))
            m.conv3p(y1, pack.wt[f"s{si}r{ri}b"], pack.bias[f"s{si}r{ri}b"],
                     x, empty, out, M, hout, hout, True, True, 1)
            if want_stash:
                st[f"s{si}r{ri}x"] = x
                st[f"s{si}r{ri}y1"] = y1
            x = out
    if want_stash:
        st["s_out"] = x                                # (M, 13, 13, 32)
    flat = m.pad2dense(x, M, 11, 11, True)             # (M, 3872) bf16
    latent = m.gemm_bias_act(flat, pack.wft, pack.bf, 1, False)
    if want_stash:
        st["flat"] = flat
    return latent, st


def encoder_bwd(m, pack: ImpalaPack, st: dict, dlat: torch.Tensor,
                latent: torch.Tensor):
    """dlat: (M, hidden) bf16 -> conv/fc gradients, accumulated STRAIGHT
    into the encoder modules' pre-zeroed .grad views (torch layout in the
    wgrad epilogues — no permute/copy kernels)."""
    ar = pack.arena
    ar.reset("bwd")
    M = dlat.shape[0]
    empty = torch.Tensor()
    enc = pack.enc
    convs = {}
    for si, stage in enumerate(enc.stages):
        convs[f"s{si}c"] = stage.conv
        for ri, res in enumerate((stage.res1, stage.res2)):
            convs[f"s{si}r{ri}a"] = res.conv1
            convs[f"s{si}r{ri}b"] = res.conv2

    # fc (+ its relu) then the flatten relu back onto the padded grid
    dflat = m.gemm_dgrad(dlat, latent, pack.wf_kn, True)
    dWf, dbf = m.gemm_wgrad(dlat, latent, st["flat"], True, True)
    hd = pack.wft.shape[0]
    enc.fc.weight.grad.copy_(
        dWf.view(hd, 11, 11, 32).permute(0, 3, 1, 2).reshape(hd, 3872))
    enc.fc.bias.grad.copy_(dbf)

    dx = ar.get("bwd", (M, 13, 13, 32))
    m.dense2pad_mask(dflat, st["s_out"], dx, M, 11, 11)

    def wgrad(name, dY, inp, H, relu_in, cin):
        # packed-layout accumulation + permute copy (torch-layout atomics
        # scatter wave lanes across cachelines — measured 4-5x slower)
        conv = convs[name]
        cout = conv.weight.shape[0]
        dWt, db = m.conv3p_wgrad(dY, inp, M, H, H, relu_in)
        conv.weight.grad.copy_(self_conv_grad(dWt, cout, cin))
        conv.bias.grad.copy_(db)

    for si in (2, 1, 0):
        hin, hout = STAGES[si]
        c = CHANS[si]
        for ri in (1, 0):
            x, y1 = st[f"s{si}r{ri}x"], st[f"s{si}r{ri}y1"]
            nb, nm = f"s{si}r{ri}b", f"s{si}r{ri}a"
            wgrad(nb, dx, y1, hout, True, c)
            dy1 = ar.get("bwd", (M, hout + 2, hout + 2, c))
            m.conv3p(dx, pack.wd[nb], empty, empty, y1, dy1,
                     M, hout, hout, False, False, 2)
            wgrad(nm, dy1, x, hout, True, c)
            dx_new = ar.get("bwd", (M, hout + 2, hout + 2, c))
            m.conv3p(dy1, pack.wd[nm], empty, dx, x, dx_new,
                     M, hout, hout, False, False, 3)
            dx = dx_new
        # maxpool backward: dx (pooled grid) -> conv-out grid
        dconv = ar.get("bwd", (M, hin + 2, hin + 2, c))
        m.maxpool3s2_bwd(dx, st[f"s{si}arg"], dconv, M, hin, hin, hout, hout)
        # stage conv wgrad (+ dgrad, except stage 0 whose input is data)
        sin = st[f"s{si}c_in"]
        cin = 4 if si == 0 else CHANS[si - 1]
        wgrad(f"s{si}c", dconv, sin, hin, False, cin)
        if si > 0:
            dx = ar.get("bwd", (M, hin + 2, hin + 2, CHANS[si - 1]))
            m.conv3p(dconv, pack.wd[f"s{si}c"], empty, empty, empty, dx,
                     M, hin, hin, False, False, 0)


def self_conv_grad(dWt: torch.Tensor, cout: int, cin: int) -> torch.Tensor:
    """(COUT, 9*cp) f32 -> (COUT, CIN, 3, 3) torch conv-weight layout."""
    cp = dWt.shape[1] // 9
    return (dWt.view(cout, 3, 3, cp)[:, :, :, :cin]
            .permute(0, 3, 1, 2).contiguous())
