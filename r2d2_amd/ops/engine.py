"""The MI355X training engine: runs the full R2D2 learner update through the
hand-written gfx950 HIP kernels with a manual (graph-free) backward pass.

Forward per update (fused): ONE online pass over the full
sequence (serving both learning- and target-position Q, replacing the
reference's two online passes, worker.py:346+352) and one target-net pass —
conv1..3 (implicit-GEMM MFMA, NHWC, fused u8 dequant) -> FC -> input GEMM ->
persistent fused LSTM (both networks in one launch) -> gathered dueling
heads -> fused double-Q/rescale/Huber loss + on-device priorities.

Backward: fused loss emits dQ; dueling-combine bwd; head GEMM d/wgrads;
scatter-add into dH_ext; persistent BPTT kernel -> dgates; LSTM weight grads
as bulk GEMMs over B*T rows; FC + conv d/wgrads (tap-class dgrad).  All
gradients land in the nn.Module parameters' .grad (f32), so the existing
clip + Adam + DDP all-reduce path applies unchanged.

Weights are prepacked bf16 (transposed/permuted per kernel contract) after
every optimizer step (`refresh_online`), and on target-net sync
(`refresh_target`).
"""

import os

import numpy as np
import torch

from .. import config as cfg
from ..models.encoders import ImpalaCNN
from . import hip_ops
from . import impala as impala_ops

PAD_HEAD = 32   # padded output width for the A-dim and V-dim head GEMMs
KPAD = 32       # LSTM input features padded to a multiple of 32


def _round_up(x, m):
    return (x + m - 1) // m * m


def _pack_items(pack, prefix="", out=None):
    """(name, holder, attr, dict_key) for every floating cuda tensor in a
    pack — same traversal order as parallel.weight_bus.pack_tensors."""
    out = [] if out is None else out
    for name in sorted(vars(pack)):
        if name.startswith("_"):
            continue
        v = vars(pack)[name]
        p = f"{prefix}{name}"
        if torch.is_tensor(v):
            if v.is_cuda and v.is_floating_point():
                out.append((p, pack, name, None))
        elif isinstance(v, dict):
            for k in sorted(v, key=repr):
                t = v[k]
                if torch.is_tensor(t) and t.is_cuda and t.is_floating_point():
                    out.append((f"{p}[{k!r}]", pack, name, k))
        elif hasattr(v, "__dict__") and hasattr(v, "refresh"):
            _pack_items(v, p + ".", out)
    return out


def _item_get(holder, attr, key):
    v = getattr(holder, attr)
    return v if key is None else v[key]


def _item_set(holder, attr, key, t):
    if key is None:
        setattr(holder, attr, t)
    else:
        getattr(holder, attr)[key] = t


class _NetPack:
    """Prepacked bf16 weights for one network (online or target).

    ``refresh()`` re-derives every packed tensor from the module weights.
    When the owning net's parameters are views of the engine's flat f32
    buffer, ``enable_fast_refresh`` replaces the ~40-launch python repack
    with ONE gather kernel per dtype: the (purely linear) packing is probed
    bit-by-bit once at init to recover, for every packed element, its
    source index in the flat buffer (plus the second source for the
    b_ih + b_hh LSTM bias sum), and the pack tensors become views of two
    flat output buffers filled by ops/hip/optim_kernels.hip gather_pack.
    """

    def __init__(self, net, device, A, with_bwd: bool):
        self.net = net
        self.device = device
        self.A = A
        self.with_bwd = with_bwd
        self._fast = None
        self.refresh()

    def refresh(self):
        if self._fast is not None:
            m, flat, m1b, m2b, bufb, m1f, m2f, buff = self._fast
            m.gather_pack(flat, m1b, m2b, bufb)
            m.gather_pack(flat, m1f, m2f, buff)
            return
        self._refresh_py()

    def enable_fast_refresh(self, flat_param, m):
        """Bit-probe the packing to build index maps, then freeze the pack
        tensors as views of two flat buffers refreshed by gather_pack."""
        dev = self.device
        n = flat_param.numel()
        save = flat_param.detach().clone()
        # the only dual-source packed tensor is lstm_bias = b_ih + b_hh;
        # zero b_hh during probing so probed indices point at b_ih, and add
        # the (constant) hh offset analytically afterwards
        bih, bhh = self.net.recurrent.bias_ih_l0, self.net.recurrent.bias_hh_l0
        ih_ofs = bih.data.storage_offset()
        hh_ofs = bhh.data.storage_offset()
        hh_n = bhh.numel()

        idx_plus1 = torch.arange(1, n + 1, dtype=torch.int64, device=dev)
        acc = {}
        for b in range(24):
            bits = ((idx_plus1 >> b) & 1).float()
            bits[hh_ofs:hh_ofs + hh_n] = 0.0
            with torch.no_grad():
                flat_param.copy_(bits)
            self._refresh_py()
            for name, holder, attr, key in _pack_items(self):
                t = _item_get(holder, attr, key)
                bit = (t.detach().float() != 0).to(torch.int64).reshape(-1)
                acc[name] = acc.get(name, 0) + (bit << b)
        with torch.no_grad():
            flat_param.copy_(save)
        self._refresh_py()

        items = _pack_items(self)
        sizes = {torch.bfloat16: 0, torch.float32: 0}
        layout = []
        for name, holder, attr, key in items:
            t = _item_get(holder, attr, key)
            layout.append((name, holder, attr, key, t.dtype, t.shape,
                           sizes[t.dtype]))
            sizes[t.dtype] += t.numel()
        bufb = torch.zeros(max(1, sizes[torch.bfloat16]),
                           dtype=torch.bfloat16, device=dev)
        buff = torch.zeros(max(1, sizes[torch.float32]),
                           dtype=torch.float32, device=dev)
        m1b = torch.full((max(1, sizes[torch.bfloat16]),), -1,
                         dtype=torch.int32, device=dev)
        m2b = torch.full_like(m1b, -1)
        m1f = torch.full((max(1, sizes[torch.float32]),), -1,
                         dtype=torch.int32, device=dev)
        m2f = torch.full_like(m1f, -1)
        for name, holder, attr, key, dtype, shape, ofs in layout:
            t = _item_get(holder, attr, key)
            nel = t.numel()
            buf = bufb if dtype == torch.bfloat16 else buff
            m1 = m1b if dtype == torch.bfloat16 else m1f
            m2 = m2b if dtype == torch.bfloat16 else m2f
            buf[ofs:ofs + nel].copy_(t.detach().reshape(-1).to(dtype))
            map1 = (acc[name] - 1).to(torch.int32)
            m1[ofs:ofs + nel] = map1
            if name == "lstm_bias":
                m2[ofs:ofs + nel] = map1 + (hh_ofs - ih_ofs)
            _item_set(holder, attr, key, buf[ofs:ofs + nel].view(shape))
        self._fast = (m, flat_param, m1b, m2b, bufb, m1f, m2f, buff)

    def _refresh_py(self):
        net = self.net
        dev = self.device
        A = self.A
        enc = net.encoder
        to_bf = lambda t: t.detach().to(dev).bfloat16().contiguous()
        f32 = lambda t: t.detach().to(dev).float().contiguous()

        self.impala = isinstance(enc, ImpalaCNN)
        if self.impala:
            if hasattr(self, "imp"):
                self.imp.refresh()
            else:
                self.imp = impala_ops.ImpalaPack(enc, dev, self.with_bwd)
        else:
            # convs: (COUT, CIN, KH, KW) -> (COUT, KH*KW*CIN)
            def pack_conv(conv):
                w = conv.weight.detach()
                wt = w.permute(0, 2, 3, 1).reshape(w.shape[0], -1)
                return to_bf(wt), f32(conv.bias)

            self.w1t, self.b1 = pack_conv(enc.conv1)
            self.w2t, self.b2 = pack_conv(enc.conv2)
            self.w3t, self.b3 = pack_conv(enc.conv3)

            # FC: torch flattens NCHW (64,7,7); conv3 output flattens HWC
            wf = enc.fc.weight.detach()          # (512, 3136) over (c,h,w)
            wf_hwc = (wf.reshape(512, 64, 7, 7).permute(0, 2, 3, 1)
                      .reshape(512, 3136))
            self.wft = to_bf(wf_hwc)
            self.bf = f32(enc.fc.bias)

        # LSTM: weight_ih (4H, 512+A+1) padded to KPAD multiple
        wih = net.recurrent.weight_ih_l0.detach()
        H4, kin = wih.shape
        self.kin = kin
        self.kin_pad = _round_up(kin, KPAD)
        wih_pad = torch.zeros(H4, self.kin_pad, device=dev)
        wih_pad[:, :kin] = wih.to(dev)
        self.wih_t = wih_pad.bfloat16().contiguous()
        self.whh_t = to_bf(net.recurrent.weight_hh_l0)        # (4H, H)
        self.lstm_bias = f32(net.recurrent.bias_ih_l0
                             + net.recurrent.bias_hh_l0)      # (4H,)

        # heads (padded to PAD_HEAD output cols, zero rows beyond real dims)
        def pack_head_out(lin, rows):
            w = torch.zeros(PAD_HEAD, lin.weight.shape[1], device=dev)
            w[:rows] = lin.weight.detach().to(dev)
            b = torch.zeros(PAD_HEAD, device=dev)
            b[:rows] = lin.bias.detach().to(dev)
            return w.bfloat16().contiguous(), b.contiguous()

        adv1, adv2 = net.advantage[0], net.advantage[2]
        val1, val2 = net.value[0], net.value[2]
        self.wa1t, self.ba1 = to_bf(adv1.weight), f32(adv1.bias)
        self.wa2t, self.ba2 = pack_head_out(adv2, A)
        self.wv1t, self.bv1 = to_bf(val1.weight), f32(val1.bias)
        self.wv2t, self.bv2 = pack_head_out(val2, 1)

        if self.with_bwd:
            # dgrad prepacks (W stored (K, N))
            if not self.impala:
                self.wf_kn = self.wft.t().contiguous()         # (3136, 512)
            self.wih_kn = self.wih_t.t().contiguous()          # (kin_pad, 4H)
            self.whh_bwd = self.whh_t.t().contiguous()         # (H, 4H)
            self.wa1_kn = self.wa1t.t().contiguous()
            self.wa2_kn = self.wa2t.t().contiguous()           # (512, PAD)
            self.wv1_kn = self.wv1t.t().contiguous()
            self.wv2_kn = self.wv2t.t().contiguous()
            if self.impala:
                return
            # nature conv dgrad prepacks
            w3 = enc.conv3.weight.detach().to(dev)             # (64,64,3,3)
            w3_nhwc = w3.permute(0, 2, 3, 1)                   # (co,dy,dx,ci)
            self.w3d = (w3_nhwc.permute(3, 1, 2, 0)            # (ci,dy,dx,co)
                        .reshape(64, 9 * 64).bfloat16().contiguous())
            self.taps3 = torch.tensor(
                [[dy, dx] for dy in range(3) for dx in range(3)],
                dtype=torch.int32, device=dev)
            w2 = enc.conv2.weight.detach().to(dev)             # (64,32,4,4)
            w2_nhwc = w2.permute(0, 2, 3, 1)                   # (co,dy,dx,ci)
            self.w2d = {}
            self.taps2 = {}
            for py in range(2):
                for px in range(2):
                    tap_list = [(dy, dx) for dy in range(py, 4, 2)
                                for dx in range(px, 4, 2)]
                    wd = torch.stack([w2_nhwc[:, d, x, :] for d, x in tap_list],
                                     dim=0)                    # (t, co, ci)
                    self.w2d[(py, px)] = (wd.permute(2, 0, 1)
                                          .reshape(32, 4 * 64)
                                          .bfloat16().contiguous())
                    self.taps2[(py, px)] = torch.tensor(
                        tap_list, dtype=torch.int32, device=dev)


class HipNetworkEngine:
    def __init__(self, online_net, target_net, device, config=None):
        c = config or cfg.get()
        self.cfg = c
        assert c.encoder in ("nature", "impala") and c.hidden_dim == 512, \
            "HIP engine supports the nature/impala 512-hidden configs"
        assert tuple(c.obs_shape[1:]) == (84, 84)
        self.C = c.obs_shape[0]
        assert self.C == 4, "conv1 kernel instantiated for 4 input channels"
        self.impala = c.encoder == "impala"
        self.A = c.action_dim
        self.H = 512
        self.device = torch.device(device)
        self.m = hip_ops.ext(required=True)
        self.online = _NetPack(online_net, self.device, self.A, with_bwd=True)
        self.target = _NetPack(target_net, self.device, self.A, with_bwd=False)
        self.online_net = online_net
        self.target_net = target_net
        self.bar = torch.zeros(512, dtype=torch.int32, device=self.device)
        self._empty = torch.Tensor()
        # persistent-LSTM watchdog: the kernels poison GridBar.poison (int32
        # word 256 of `bar`) on a bounded-spin timeout and return with
        # partially written outputs.  The host snapshots that word right
        # after each launch (async, no sync) and raises at the NEXT step —
        # a poisoned run must never silently corrupt gradients.
        self._poison_pin = torch.zeros(2, dtype=torch.int32, pin_memory=True)

        # flat parameter/grad/Adam-moment buffers: module params become views
        # so the whole network is clipped + stepped by TWO kernels
        # (ops/hip/optim_kernels.hip) and DP all-reduce is ONE collective.
        params = [p for p in online_net.parameters()]
        n_total = sum(p.numel() for p in params)
        dev = self.device
        self.flat_param = torch.empty(n_total, dtype=torch.float32, device=dev)
        self.flat_grad = torch.zeros(n_total, dtype=torch.float32, device=dev)
        self.exp_avg = torch.zeros(n_total, dtype=torch.float32, device=dev)
        self.exp_avg_sq = torch.zeros(n_total, dtype=torch.float32, device=dev)
        self.norm_buf = torch.zeros(1, dtype=torch.float32, device=dev)
        self.adam_t = 0
        # contiguous flat-grad ranges per backward stage, in COMPLETION
        # order (heads -> lstm -> encoder).  The DP learner launches each
        # segment's RCCL all-reduce as soon as the manual backward finishes
        # filling it, overlapping communication with the remaining backward
        # (the comm runs on RCCL's stream; later compute keeps the default
        # stream busy).
        named = list(online_net.named_parameters())
        seg_of = lambda name: ("heads" if name.split(".")[0]
                               in ("advantage", "value")
                               else "lstm" if name.startswith("recurrent")
                               else "encoder")
        self.seg_ranges = {}
        ofs = 0
        for (name, _), p in zip(named, params):
            n = p.numel()
            self.flat_param[ofs:ofs + n].copy_(p.data.to(dev).view(-1))
            p.data = self.flat_param[ofs:ofs + n].view(p.shape)
            p.grad = self.flat_grad[ofs:ofs + n].view(p.shape)
            s = seg_of(name)
            lo, hi = self.seg_ranges.get(s, (ofs, ofs))
            assert hi == ofs, "segment params must be flat-contiguous"
            self.seg_ranges[s] = (min(lo, ofs), ofs + n)
            ofs += n
        self.timing = bool(os.environ.get("R2D2_ENGINE_TIMING"))
        self._events = []
        # per-step repack -> ONE gather kernel per dtype (maps bit-probed
        # from the now-flat param buffer; see _NetPack.enable_fast_refresh)
        self.online.enable_fast_refresh(self.flat_param, self.m)
        # hipGraph capture of the fwd+bwd launch sequence — MEASURED
        # NEGATIVE, default off (R2D2_HIP_GRAPH=1 to enable):
        # - prebuilt batches: 11.41k vs 11.54k seq/s uncaptured (the ~55
        #   launches are already async-submitted; replay saves nothing);
        # - with the GPU-resident replay in the loop it collapses to ~2.8k:
        #   the graph's private memory pool evicts the caching allocator's
        #   blocks for the per-sample gather outputs, degenerating into
        #   hipMalloc churn.
        # Kept (capture-correct, all tests pass with it on) for future
        # launch-bound configs; finding a real ROCm defect on the way: a
        # captured hipMemsetAsync replays a garbage fill value from the
        # 2nd replay (gpurun_out/graph_probe2.log) — the LSTM workspace
        # reset is a plain kernel now (zero_gridbar_kernel).
        self._graph = None
        self._graph_key = None
        self._use_graph = (os.environ.get("R2D2_HIP_GRAPH", "0") == "1"
                           and not self.timing)

    def _mark(self, name):
        if self.timing:
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self._events.append((name, ev))

    def timing_report(self):
        torch.cuda.synchronize()
        out = []
        for (n0, e0), (n1, e1) in zip(self._events, self._events[1:]):
            out.append((n1, e0.elapsed_time(e1)))
        self._events.clear()
        return out

    def refresh_online(self):
        self.online.refresh()

    def optimizer_step(self, lr, eps, max_norm, betas=(0.9, 0.999)):
        """Fused grad-norm clip + Adam over the flat buffers (K12/K13)."""
        self.adam_t += 1
        self.m.grad_sumsq(self.flat_grad, self.norm_buf)
        self.m.adam_step(self.flat_param, self.flat_grad, self.exp_avg,
                         self.exp_avg_sq, self.norm_buf, max_norm, lr,
                         betas[0], betas[1], eps, self.adam_t)

    def refresh_target(self):
        self.target.refresh()

    # ------------------------------------------------------------------
    _fwd_band = os.environ.get("R2D2_CONV_FWD_BAND", "1") != "0"

    def _encoder_fwd(self, pack, obs_hwc_u8, want_stash=True):
        """obs: (M, 84, 84, C) uint8 -> latent (M, 512) bf16 + stashes."""
        m = self.m
        if self.impala:
            return impala_ops.encoder_fwd(m, pack.imp, obs_hwc_u8, want_stash,
                                          xp=getattr(self, "_xp", None))
        M = obs_hwc_u8.shape[0]
        if self._fwd_band:
            # per-image band forwards: image + weights LDS-resident, one
            # global read/dequant per input element
            a1 = m.conv_fwd_band(obs_hwc_u8, pack.w1t, pack.b1, 1, M)
            a2 = m.conv_fwd_band(a1, pack.w2t, pack.b2, 2, M)
            a3 = m.conv_fwd_band(a2, pack.w3t, pack.b3, 3, M)
        else:
            a1 = m.conv_fwd(obs_hwc_u8, pack.w1t, pack.b1, 1, M, 84, 84,
                            20, 20, True)
            a2 = m.conv_fwd(a1, pack.w2t, pack.b2, 2, M, 20, 20, 9, 9, True)
            a3 = m.conv_fwd(a2, pack.w3t, pack.b3, 3, M, 9, 9, 7, 7, True)
        flat = a3.view(M, 3136)
        latent = m.gemm_bias_act(flat, pack.wft, pack.bf, 1, False)
        return latent, (a1, a2, a3, flat)

    def _lstm_input(self, pack, latent, last_action, last_reward):
        """Build padded rin (one fused kernel) and the input-GEMM X."""
        M = latent.shape[0]
        rin = self.m.assemble_rin(
            latent.contiguous(),
            last_action.reshape(M, self.A).float().contiguous(),
            last_reward.reshape(M).float().contiguous(), pack.kin_pad)
        X = self.m.gemm_bias_act(rin, pack.wih_t, pack.lstm_bias, 0, False)
        return rin, X

    def _heads_fwd(self, pack, h):
        """h: (R, 512) bf16 -> q (R, A) f32 + head stashes."""
        m = self.m
        adv1 = m.gemm_bias_act(h, pack.wa1t, pack.ba1, 1, False)
        adv2 = m.gemm_bias_act(adv1, pack.wa2t, pack.ba2, 0, True)
        val1 = m.gemm_bias_act(h, pack.wv1t, pack.bv1, 1, False)
        val2 = m.gemm_bias_act(val1, pack.wv2t, pack.bv2, 0, True)
        q = m.dueling_combine(adv2, val2, self.A)
        return q, (adv1, val1)

    # ------------------------------------------------------------------
    _pos_cache = None

    def _positions(self, burn, learn, fwd, T):
        """Host-side gather indices.  Returns (learn_pos, tgt_pos) flat int64
        arrays indexing (b * (T+1) + t + 1) rows of Hout, plus per-row b,t
        (for the dHext scatter).  Cached by batch layout — fixed-shape
        batches (the common case) skip the python loop and re-upload."""
        key = (T, burn.numpy().tobytes(), learn.numpy().tobytes(),
               fwd.numpy().tobytes())
        if self._pos_cache is None:
            self._pos_cache = {}
        if key in self._pos_cache:
            return self._pos_cache[key]
        n = self.cfg.forward_steps
        bu = burn.numpy().astype(np.int64)
        le = learn.numpy().astype(np.int64)
        fw = fwd.numpy().astype(np.int64)
        R = int(le.sum())
        # vectorized ragged expansion (no per-sample python loop)
        b_of = np.repeat(np.arange(len(le), dtype=np.int64), le)
        starts = np.zeros(len(le), dtype=np.int64)
        starts[1:] = np.cumsum(le[:-1])
        within = np.arange(R, dtype=np.int64) - np.repeat(starts, le)
        t_learn = bu[b_of] + within
        t_tgt = np.minimum(t_learn + n, (bu + le + fw - 1)[b_of])
        dev = self.device
        lp = b_of * (T + 1) + t_learn + 1
        tp = b_of * (T + 1) + t_tgt + 1
        lbt = b_of * T + t_learn
        lp_t = torch.from_numpy(lp).to(dev)
        tp_t = torch.from_numpy(tp).to(dev)
        # inverse map for the dHext scatter kernel: (b*T + t) -> learn row
        # index, -1 where no learning position sits (learn t's are unique
        # per sample, so the map is collision-free)
        row_of = np.full(len(le) * T, -1, dtype=np.int32)
        row_of[lbt] = np.arange(R, dtype=np.int32)
        row_of_t = torch.from_numpy(row_of).to(dev)
        lens_dev = (burn + learn + fwd).to(torch.int32).to(dev)
        seg = torch.zeros(len(burn) + 1, dtype=torch.int32)
        seg[1:] = torch.cumsum(learn.to(torch.int32), 0)
        out = (lp_t, tp_t, row_of_t, lens_dev, seg.to(dev))
        if len(self._pos_cache) > 64:   # bounded (fixed layouts in practice)
            self._pos_cache.clear()
        self._pos_cache[key] = out
        return out

    # ------------------------------------------------------------------
    # -- hipGraph capture ----------------------------------------------
    def _layout_key(self, batch):
        return (tuple(batch.obs.shape), batch.obs.dtype,
                batch.burn_in_steps.numpy().tobytes(),
                batch.learning_steps.numpy().tobytes(),
                batch.forward_steps.numpy().tobytes())

    def _copy_into_static(self, batch):
        sb = self._static_batch
        for f in ("obs", "last_action", "last_reward", "hidden", "action",
                  "n_step_reward", "gamma", "is_weights"):
            getattr(sb, f).copy_(getattr(batch, f), non_blocking=True)

    def _try_capture(self, batch, key):
        import types
        dev = self.device
        sb = types.SimpleNamespace(
            obs=batch.obs.to(dev).contiguous().clone(),
            last_action=batch.last_action.to(dev).contiguous().clone(),
            last_reward=batch.last_reward.to(dev).contiguous().clone(),
            hidden=batch.hidden.to(dev).contiguous().clone(),
            action=batch.action.to(dev).contiguous().clone(),
            n_step_reward=batch.n_step_reward.to(dev).contiguous().clone(),
            gamma=batch.gamma.to(dev).contiguous().clone(),
            is_weights=batch.is_weights.to(dev).contiguous().clone(),
            burn_in_steps=batch.burn_in_steps,
            learning_steps=batch.learning_steps,
            forward_steps=batch.forward_steps)
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):                  # allocator warmup
                    self._train_step_impl(sb)
            torch.cuda.current_stream().wait_stream(s)
            g = torch.cuda.CUDAGraph()
            # thread_local: the replay-ingest thread keeps streaming actor
            # blocks on its own streams while we capture — global capture
            # mode would invalidate them and poison the whole HIP context
            with torch.cuda.graph(g, capture_error_mode="thread_local"):
                loss, prio = self._train_step_impl(sb)
        except Exception as e:
            print(f"[engine] hipGraph capture failed ({e!r}); "
                  f"running uncaptured", flush=True)
            self._use_graph = False
            return False
        self._graph = g
        self._graph_key = key
        self._static_batch = sb
        self._graph_out = (loss, prio)
        return True

    def train_step(self, batch, grad_hook=None):
        """Full fused update (hipGraph replay when the batch layout matches
        the captured graph).  Fills .grad on the online nn.Module params
        and returns (loss tensor, per-sequence priority tensor, both on
        device).  ``grad_hook(segment)`` fires when a flat-grad segment
        ("heads", "lstm", "encoder") is complete — the DP learner uses it
        to overlap the RCCL all-reduce with the rest of the manual
        backward (graphs are bypassed in that case so the collectives can
        interleave)."""
        if bool(self._poison_pin.any()):
            raise RuntimeError(
                "persistent LSTM kernel poisoned on a previous step "
                "(bounded-spin timeout) — outputs/gradients of that step "
                "are invalid; aborting instead of training on them")
        if self._use_graph and grad_hook is None:
            key = self._layout_key(batch)
            if self._graph_key != key:
                self._try_capture(batch, key)
            if self._graph_key == key:
                # capture only RECORDS the work — every step (including the
                # capture step) must replay to actually execute
                self._copy_into_static(batch)
                self._graph.replay()
                return self._graph_out
        return self._train_step_impl(batch, grad_hook)

    def _train_step_impl(self, batch, grad_hook=None):
        """The uncaptured launch sequence (also the capture body)."""
        m = self.m
        c = self.cfg
        dev = self.device
        B, T = batch.obs.shape[:2]
        A, H = self.A, self.H

        # ---- inputs ----------------------------------------------------
        obs = batch.obs
        if obs.shape[-1] == self.C and obs.shape[2] == 84:   # already HWC
            obs_hwc = obs.reshape(B * T, 84, 84, self.C)
        else:                                      # (B,T,C,84,84) -> NHWC
            obs_hwc = obs.permute(0, 1, 3, 4, 2).reshape(B * T, 84, 84, self.C)
            obs_hwc = obs_hwc.contiguous()
        la = batch.last_action.to(dev)
        lr = batch.last_reward.to(dev)
        md = getattr(batch, "meta_dev", None)
        if md is not None:
            # replay-sampled batches: position arrays computed on-device
            # from the sample metadata (one kernel; every such batch is
            # ragged, so the host cache below would miss every step)
            lp_t, tp_t, row_of, lens = m.positions_meta(
                md, batch.seg_dev, T, c.forward_steps,
                batch.action.shape[0], int(c.learning_steps))
            seg = batch.seg_dev
        else:
            lp_t, tp_t, row_of, lens, seg = self._positions(
                batch.burn_in_steps, batch.learning_steps,
                batch.forward_steps, T)
        R = lp_t.shape[0]
        init = batch.hidden.float().contiguous()   # (2, B, H)

        self._mark("start")
        # ---- forward ---------------------------------------------------
        if self.impala:
            # pack frames once; online and target share the padded u8 tensor
            self._xp = impala_ops.pack_obs(m, self.online.imp, obs_hwc)
        lat_o, enc_stash = self._encoder_fwd(self.online, obs_hwc)
        self._mark("enc_online")
        lat_t, _ = self._encoder_fwd(self.target, obs_hwc, want_stash=False)
        self._mark("enc_target")
        rin_o, X_o = self._lstm_input(self.online, lat_o, la, lr)
        _, X_t = self._lstm_input(self.target, lat_t, la, lr)
        self._mark("lstm_input")
        Xo = X_o.view(B, T, 4 * H)
        Xt = X_t.view(B, T, 4 * H)
        Ho, Co, Ht, Ct, stash = m.lstm_fwd(
            Xo, Xt, self.online.whh_t, self.target.whh_t,
            init, init, lens, self.bar, True)
        self._poison_pin[0:1].copy_(self.bar[256:257], non_blocking=True)
        self._mark("lstm_fwd")

        Ho_flat = Ho.view(-1, H)
        h_learn = Ho_flat.index_select(0, lp_t)
        h_tgt_o = Ho_flat.index_select(0, tp_t)
        h_tgt_t = Ht.view(-1, H).index_select(0, tp_t)

        # online heads on [learn; tgt] rows in one pass
        h_cat = torch.cat([h_learn, h_tgt_o], 0)
        q_cat, (adv1_o, val1_o) = self._heads_fwd(self.online, h_cat)
        q_learn, q_online_tgt = q_cat[:R], q_cat[R:]
        q_tgt, _ = self._heads_fwd(self.target, h_tgt_t)
        self._mark("heads")

        # ---- fused loss + priorities ----------------------------------
        loss, dq, abs_td, _ = m.fused_double_q_loss(
            q_learn.contiguous(), q_online_tgt.contiguous(), q_tgt.contiguous(),
            batch.action.view(-1).long(), batch.n_step_reward, batch.gamma,
            batch.is_weights, c.rescale_eps, c.huber_kappa,
            0 if c.loss_fn == "mse" else 1)
        prio = m.segment_priority(abs_td, seg, c.prio_eta)
        self._mark("loss")

        # ---- backward --------------------------------------------------
        # weight grads accumulate STRAIGHT into the flat .grad views
        # (pre-zeroed once) — no per-parameter copy/permute kernels.
        # Only the R learning rows carry gradient (the target-position rows
        # of q_cat enter the loss detached, reference worker.py:346), so the
        # whole head backward runs on R rows, not 2R.
        ON = self.online
        net = self.online_net
        self.flat_grad.zero_()
        adv1_l, val1_l, h_l = adv1_o[:R], val1_o[:R], h_cat[:R]
        dadv2, dval2 = m.dueling_combine_bwd(dq.contiguous(),
                                             PAD_HEAD, PAD_HEAD)
        self._mark("hb_duel")
        # adv path
        dadv1 = m.gemm_dgrad(dadv2, self._empty, ON.wa2_kn, False)
        dh_a = m.gemm_dgrad(dadv1, adv1_l, ON.wa1_kn, True)
        # value path
        dval1 = m.gemm_dgrad(dval2, self._empty, ON.wv2_kn, False)
        dh_v = m.gemm_dgrad(dval1, val1_l, ON.wv1_kn, True)
        self._mark("hb_dgrads")
        m.gemm_wgrad_into(dadv2, self._empty, adv1_l, False,
                          net.advantage[2].weight.grad,
                          net.advantage[2].bias.grad)
        m.gemm_wgrad_into(dadv1, adv1_l, h_l, True,
                          net.advantage[0].weight.grad,
                          net.advantage[0].bias.grad)
        m.gemm_wgrad_into(dval2, self._empty, val1_l, False,
                          net.value[2].weight.grad, net.value[2].bias.grad)
        m.gemm_wgrad_into(dval1, val1_l, h_l, True,
                          net.value[0].weight.grad, net.value[0].bias.grad)
        if grad_hook is not None:
            grad_hook("heads")
        self._mark("hb_wgrads")

        # dHext (B, T, H): dh_a+dh_v scattered to the learning positions in
        # one kernel via the cached inverse map (no zeros fill / index_add_)
        dHext = m.scatter_dh(dh_a, dh_v, row_of, B * T).view(B, T, H)
        self._mark("head_bwd")

        dgates = m.lstm_bwd(stash, Co, Ho, dHext.contiguous(), ON.whh_bwd,
                            lens, self.bar)
        self._poison_pin[1:2].copy_(self.bar[256:257], non_blocking=True)
        self._mark("lstm_bwd")
        dgates_flat = dgates.view(B * T, 4 * H)

        h_prev = Ho[:, :T].reshape(B * T, H).contiguous()
        m.gemm_wgrad_into(dgates_flat, self._empty, h_prev, False,
                          net.recurrent.weight_hh_l0.grad, self._empty)
        m.gemm_wgrad_into(dgates_flat, self._empty, rin_o, False,
                          net.recurrent.weight_ih_l0.grad,
                          net.recurrent.bias_ih_l0.grad)
        net.recurrent.bias_hh_l0.grad.copy_(net.recurrent.bias_ih_l0.grad)
        if grad_hook is not None:
            grad_hook("lstm")
        # only the latent slice of the padded LSTM-input grad is needed
        dlat = m.gemm_dgrad(dgates_flat, self._empty, ON.wih_kn, False, 512)
        self._mark("lstm_wgrads")

        M = B * T
        if self.impala:
            impala_ops.encoder_bwd(m, ON.imp, enc_stash, dlat, lat_o)
            self._mark("conv_bwd")
            if grad_hook is not None:
                grad_hook("encoder")
            self._mark("grad_write")
            return loss.squeeze(0), prio

        a1, a2, a3, flat = enc_stash
        lat_bf = lat_o  # forward output (relu mask source)
        # FC dgrad with BOTH relu backwards fused: the mask of the FC's own
        # relu (lat) on load, conv3's relu mask (a3) on store — dflat leaves
        # this kernel pre-masked, so no elementwise mask/pad passes run on
        # the conv gradients at all.
        dflat = m.gemm_dgrad(dlat, lat_bf, ON.wf_kn, True, 0, a3)
        dWf, dbf = m.gemm_wgrad(dlat, lat_bf, flat, True, True)

        # conv3 backward.  Conv weight grads accumulate in the packed
        # (COUT, ky, kx, c) layout — torch-layout atomics scatter each
        # 16-lane wave across 16 cachelines (measured 4-5x slower) — and
        # are permuted into .grad afterwards (tiny tensors).
        self._mark("fc_bwd")
        dflat3 = dflat.view(M * 49, 64)   # pre-masked by a3
        # per-image band wgrads: whole input image staged in LDS once
        # (single dequant per element; conv1 patch traffic 557 -> 154 MB)
        dW3, db3 = m.conv_wgrad_band(dflat3, a3, a2, 3, M)
        # dense bounds-checked dgrad (no zero-padded staging copies); the
        # output mask fuses conv2's relu backward into the d_a2 store
        d_a2 = torch.empty(M, 9, 9, 64, device=dev, dtype=torch.bfloat16)
        m.conv_dgrad_dense(dflat3, ON.w3d, ON.taps3, a2,
                           M, 7, 7, 64, 9, 9, 64, 0, 0, 1, d_a2)
        # conv2 backward (d_a2 pre-masked by a2)
        d_a2f = d_a2.view(M * 81, 64)
        dW2, db2 = m.conv_wgrad_band(d_a2f, a2, a1, 2, M)
        d_a1 = torch.empty(M, 20, 20, 32, device=dev, dtype=torch.bfloat16)
        for py in range(2):
            for px in range(2):
                m.conv_dgrad_dense(d_a2f, ON.w2d[(py, px)],
                                   ON.taps2[(py, px)], a1,
                                   M, 9, 9, 64, 20, 20, 32, py, px, 2, d_a1)
        # conv1 wgrad (no dgrad: input is data; d_a1 pre-masked by a1).
        # Stays on the CHUNKED kernel: measured 0.434 vs 0.474 ms for the
        # per-image band variant (the 8x8 s4 patch field re-reads only 4x,
        # and the band's whole-image slab costs more occupancy than the
        # dequant re-reads save; conv2/conv3 with 9x/4x re-read DO win).
        dW1, db1 = m.conv_wgrad(d_a1.view(M * 400, 32), a1, obs_hwc, 1,
                                M, 84, 84, 20, 20, 32, 8 * 8 * self.C)
        self._mark("conv_bwd")

        enc = net.encoder

        def conv_grad(dwt, cout, kh, kw, cin):
            return dwt.view(cout, kh, kw, cin).permute(0, 3, 1, 2)

        enc.conv1.weight.grad.copy_(conv_grad(dW1, 32, 8, 8, self.C))
        enc.conv1.bias.grad.copy_(db1)
        enc.conv2.weight.grad.copy_(conv_grad(dW2, 64, 4, 4, 32))
        enc.conv2.bias.grad.copy_(db2)
        enc.conv3.weight.grad.copy_(conv_grad(dW3, 64, 3, 3, 64))
        enc.conv3.bias.grad.copy_(db3)
        enc.fc.weight.grad.copy_(
            dWf.view(512, 7, 7, 64).permute(0, 3, 1, 2).reshape(512, 3136))
        enc.fc.bias.grad.copy_(dbf)
        if grad_hook is not None:
            grad_hook("encoder")

        self._mark("grad_write")
        return loss.squeeze(0), prio


class HipInference:
    """K15: the single-step actor-inference fast path through the gfx950
    kernels (SURVEY §2.3 K15 — the reference runs this on CPU per actor,
    model.py:65-79).  Used by VectorActor on a cuda device: one batched
    encoder pass + one LSTM cell step + the dueling heads per tick, with
    prepacked bf16 weights refreshed on each weight pull."""

    def __init__(self, net, device, config=None):
        c = config or cfg.get()
        assert c.encoder in ("nature", "impala") and c.hidden_dim == 512
        assert tuple(c.obs_shape) == (4, 84, 84)
        self.cfg = c
        self.A = net.action_dim
        self.device = torch.device(device)
        self.m = hip_ops.ext(required=True)
        self.net = net
        self.pack = _NetPack(net, self.device, self.A, with_bwd=False)
        self.impala = self.pack.impala

    def refresh(self):
        self.pack.refresh()

    @torch.no_grad()
    def forward(self, obs_u8_nchw, last_action, last_reward, hidden):
        """obs: (E, 4, 84, 84) u8 on device; hidden: (h, c) each (1, E, 512)
        f32.  Returns (q (E, A) f32, (h', c'))."""
        m = self.m
        E = obs_u8_nchw.shape[0]
        obs_hwc = obs_u8_nchw.permute(0, 2, 3, 1).contiguous()
        if self.impala:
            lat, _ = impala_ops.encoder_fwd(m, self.pack.imp, obs_hwc, False)
        else:
            p = self.pack
            a1 = m.conv_fwd(obs_hwc, p.w1t, p.b1, 1, E, 84, 84, 20, 20, True)
            a2 = m.conv_fwd(a1, p.w2t, p.b2, 2, E, 20, 20, 9, 9, True)
            a3 = m.conv_fwd(a2, p.w3t, p.b3, 3, E, 9, 9, 7, 7, True)
            lat = m.gemm_bias_act(a3.view(E, 3136), p.wft, p.bf, 1, False)
        p = self.pack
        rin = m.assemble_rin(lat.contiguous(),
                             last_action.float().contiguous(),
                             last_reward.reshape(E).float().contiguous(),
                             p.kin_pad)
        x = m.gemm_bias_act(rin, p.wih_t, p.lstm_bias, 0, True)     # f32
        h0, c0 = hidden
        hg = m.gemm_bias_act(h0.reshape(E, 512).bfloat16().contiguous(),
                             p.whh_t, torch.Tensor(), 0, True)      # f32
        gates = (x + hg).view(E, 4, 512)
        i_ = torch.sigmoid(gates[:, 0])
        f_ = torch.sigmoid(gates[:, 1])
        g_ = torch.tanh(gates[:, 2])
        o_ = torch.sigmoid(gates[:, 3])
        c1 = f_ * c0.reshape(E, 512) + i_ * g_
        h1 = o_ * torch.tanh(c1)
        hb = h1.bfloat16().contiguous()
        adv1 = m.gemm_bias_act(hb, p.wa1t, p.ba1, 1, False)
        adv2 = m.gemm_bias_act(adv1, p.wa2t, p.ba2, 0, True)
        val1 = m.gemm_bias_act(hb, p.wv1t, p.bv1, 1, False)
        val2 = m.gemm_bias_act(val1, p.wv2t, p.bv2, 0, True)
        q = m.dueling_combine(adv2, val2, self.A)
        return q, (h1.view(1, E, 512), c1.view(1, E, 512))
