"""GPU-resident prioritized replay: HBM block store + device sum-tree +
on-device batch assembly.

The MI355X replacement for the reference's host-RAM ReplayBuffer + CPU
PriorityTree + Python batch assembler: 288 GB of
HBM3E holds the whole 4M-transition store (uint8 frames), sampling descends
the tree on-device, the padded training batch is gathered straight out of
the block store, and priority updates (with the ring-overwrite stale mask of
reference worker.py:247-256) never leave the GPU.

Ingest takes host `Block`s (from CPU actors) through pinned staging buffers
on a dedicated copy stream.
"""

from typing import Optional

import numpy as np
import torch

from .. import config as cfg
from ..ops import hip_ops


class GpuReplayBuffer:
    def __init__(self, device="cuda", capacity: Optional[int] = None,
                 batch_size: Optional[int] = None):
        c = cfg.get()
        self.cfg = c
        self.device = torch.device(device)
        assert len(c.obs_shape) == 3, "GPU replay stores uint8 frames"
        self.obs_shape = tuple(c.obs_shape)
        self.frame_bytes = int(np.prod(self.obs_shape))
        self.A = c.action_dim
        self.H = c.hidden_dim
        self.alpha = c.prio_exponent
        self.beta = c.importance_sampling_exponent
        self.eta = c.prio_eta
        self.batch_size = batch_size or c.batch_size

        self.block_len = c.block_length
        self.learn_len = c.learning_steps
        self.spb = c.seq_per_block
        self.capacity = capacity or c.buffer_capacity
        self.num_blocks = self.capacity // self.block_len
        self.num_sequences = self.num_blocks * self.spb
        self.obs_rows = c.burn_in_steps + self.block_len + 1
        self.T = c.seq_len  # fixed padded length (burn+learn+fwd)

        dev = self.device
        nb, spb = self.num_blocks, self.spb
        self.obs_store = torch.zeros(nb, self.obs_rows, self.frame_bytes,
                                     dtype=torch.uint8, device=dev)
        self.la_store = torch.zeros(nb, self.obs_rows, dtype=torch.uint8, device=dev)
        self.lr_store = torch.zeros(nb, self.obs_rows, dtype=torch.float32, device=dev)
        self.act_store = torch.zeros(nb, self.block_len, dtype=torch.uint8, device=dev)
        self.nsr_store = torch.zeros(nb, self.block_len, dtype=torch.float32, device=dev)
        self.gam_store = torch.zeros(nb, self.block_len, dtype=torch.float32, device=dev)
        self.hid_store = torch.zeros(nb * spb, 2 * self.H, dtype=torch.float32,
                                     device=dev)
        zi = lambda: torch.zeros(nb * spb, dtype=torch.int32, device=dev)
        self.burn_s, self.learn_s, self.fwd_s = zi(), zi(), zi()
        self.obs_start_s, self.learn_off_s = zi(), zi()

        # heap-layout f64 sum-tree
        self.num_levels = 1
        while (1 << self.num_levels) < self.num_sequences:
            self.num_levels += 1
        self.leaf_offset = (1 << self.num_levels) - 1
        self.tree = torch.zeros(2 * (1 << self.num_levels) - 1,
                                dtype=torch.float64, device=dev)

        self.block_ptr = 0
        self.blocks_written = 0   # min(total ingested, num_blocks)
        self.size = 0
        self.env_steps = 0
        self._ext = hip_ops.ext(required=True)
        # pinned staging + dedicated copy stream for the frame payload
        # (actor->learner trajectory push via pinned host
        # buffers and hipMemcpyAsync on a copy stream, overlapped with
        # compute on the default stream)
        self._copy_stream = torch.cuda.Stream(device=dev)
        # rotating pinned staging buffers: the host only waits for the
        # buffer N-ago to drain (no per-block stream sync), and ALL ingest
        # device work runs on the copy stream — the training stream never
        # waits on ingest (the sample stream fences on _ev_ingest instead)
        self._n_pin = 4
        self._pin_obs = [torch.empty(self.obs_rows, self.obs_store.shape[-1],
                                     dtype=torch.uint8, pin_memory=True)
                         for _ in range(self._n_pin)]
        self._pin_ev = [torch.cuda.Event() for _ in range(self._n_pin)]
        self._pin_used = [False] * self._n_pin
        self._pin_cursor = 0
        self._ev_ingest = torch.cuda.Event()
        # side stream for prefetch sampling (sample_async): sampling of the
        # NEXT batch overlaps the current training step.  Tree reads and
        # writes are fenced both ways: the sample stream waits on the last
        # tree write (_ev_tree), and tree writers wait on the last sample's
        # reads (_ev_sampled) — one step of priority staleness, well inside
        # the reference's own <=12-batch staleness.
        self._sample_stream = torch.cuda.Stream(device=dev)
        self._ev_tree = torch.cuda.Event()
        self._ev_sampled = torch.cuda.Event()
        # rotating pinned landing buffers for the small per-sample metadata
        # D2H (a pageable destination would silently make the copy — and
        # the host — block inside sample_async)
        B0 = self.batch_size
        self._pin_meta = [torch.empty(3, B0, dtype=torch.int32,
                                      pin_memory=True) for _ in range(2)]
        self._pin_seg = [torch.empty(B0 + 1, dtype=torch.int32,
                                     pin_memory=True) for _ in range(2)]
        self._pin_i = 0

    # ------------------------------------------------------------------
    def __len__(self):
        return self.size

    @property
    def total_priority(self) -> float:
        return float(self.tree[0].item())

    def ingest(self, block, priorities: np.ndarray):
        """Copy one actor Block into the device store and set its sequence
        priorities.  `block` is a worker.Block (numpy); `priorities` has
        seq_per_block entries, zero-padded for unused slots."""
        slot = self.block_ptr
        rows = block.obs.shape[0]
        steps = block.action.shape[0]
        nseq = block.num_sequences
        dev = self.device

        # store frames HWC (channels innermost) — the conv kernels are NHWC,
        # so gathered batches feed conv1 with no device-side permute
        obs_np = block.obs
        if obs_np.ndim == 4:
            obs_np = np.ascontiguousarray(obs_np.transpose(0, 2, 3, 1))
        obs_flat = torch.from_numpy(obs_np.reshape(rows, -1))
        assert obs_flat.dtype == torch.uint8
        # frames: host memcpy into the next rotating pinned buffer (waiting
        # only for THAT buffer's previous H2D, n_pin blocks ago), then every
        # device-side ingest op — H2D copies, metadata, tree update — runs
        # on the copy stream.  The training stream never waits on ingest;
        # sample_async fences its gathers on _ev_ingest instead.
        pi = self._pin_cursor
        self._pin_cursor = (pi + 1) % self._n_pin
        if self._pin_used[pi]:
            self._pin_ev[pi].synchronize()
        pin = self._pin_obs[pi]
        pin[:rows].copy_(obs_flat)
        self._pin_used[pi] = True

        # per-sequence metadata (host side)
        burn = block.burn_in_steps.astype(np.int32)
        learn = block.learning_steps.astype(np.int32)
        fwd = block.forward_steps.astype(np.int32)
        learn_off = np.zeros(nseq, dtype=np.int32)
        learn_off[1:] = np.cumsum(learn[:-1])
        obs_start = (slot * self.obs_rows + int(burn[0]) + learn_off).astype(np.int32)
        meta = np.zeros((5, self.spb), dtype=np.int32)
        meta[0, :nseq] = burn
        meta[1, :nseq] = learn
        meta[2, :nseq] = fwd
        meta[3, :nseq] = obs_start
        meta[4, :nseq] = learn_off
        base = slot * self.spb

        with torch.cuda.stream(self._copy_stream):
            # don't overwrite a slot an in-flight sample may be gathering
            self._copy_stream.wait_event(self._ev_sampled)
            self.obs_store[slot, :rows].copy_(pin[:rows], non_blocking=True)
            self._pin_ev[pi].record(self._copy_stream)
            la_idx = torch.from_numpy(np.ascontiguousarray(
                block.last_action.argmax(1).astype(np.uint8)))
            self.la_store[slot, :rows].copy_(la_idx, non_blocking=True)
            self.lr_store[slot, :rows].copy_(
                torch.from_numpy(block.last_reward), non_blocking=True)
            self.act_store[slot, :steps].copy_(
                torch.from_numpy(block.action), non_blocking=True)
            self.nsr_store[slot, :steps].copy_(
                torch.from_numpy(block.n_step_reward), non_blocking=True)
            self.gam_store[slot, :steps].copy_(
                torch.from_numpy(block.gamma), non_blocking=True)
            self.hid_store[base: base + nseq].copy_(
                torch.from_numpy(block.hidden.reshape(nseq, -1)),
                non_blocking=True)
            mt = torch.from_numpy(meta).to(dev, non_blocking=True)
            sl = slice(base, base + self.spb)
            self.burn_s[sl] = mt[0]
            self.learn_s[sl] = mt[1]
            self.fwd_s[sl] = mt[2]
            self.obs_start_s[sl] = mt[3]
            self.learn_off_s[sl] = mt[4]

            # priorities (zero for unused seq slots kills them in the tree)
            idxes = torch.arange(base, base + self.spb, dtype=torch.int64,
                                 device=dev)
            prio = torch.from_numpy(np.ascontiguousarray(
                priorities.astype(np.float32))).to(dev, non_blocking=True)
            self._ext.sumtree_update(self.tree, self.leaf_offset, idxes, prio,
                                     self.alpha, 0, 0, self.spb,
                                     self.num_blocks)
            self._ev_ingest.record(self._copy_stream)

        self.size += int(learn.sum())
        if self.size > self.capacity:
            self.size = min(self.size, self.capacity)
        self.env_steps += int(learn.sum())
        self.blocks_written = min(self.blocks_written + 1, self.num_blocks)
        self.block_ptr = (self.block_ptr + 1) % self.num_blocks

    # ------------------------------------------------------------------
    def sample_async(self, batch_size: Optional[int] = None):
        """Launch sampling + gather on the side stream and start the small
        D2H of per-sample lengths; returns a token for sample_wait().  Call
        right after launching the training step to overlap the two."""
        B = batch_size or self.batch_size
        dev = self.device
        self._ev_tree.record(torch.cuda.current_stream())
        with torch.cuda.stream(self._sample_stream):
            self._sample_stream.wait_event(self._ev_tree)
            # gathers must see fully-ingested blocks (frames + metadata +
            # tree), which the copy stream publishes via _ev_ingest
            self._sample_stream.wait_event(self._ev_ingest)
            jitter = torch.rand(B, device=dev)
            # fp-edge descent overshoot is clamped to the highest WRITTEN
            # leaf, not tree capacity — a partially filled ring must never
            # surface an unwritten zero-priority leaf (its floored priority
            # would explode the IS weight)
            max_leaf = min(self.blocks_written * self.spb,
                           self.num_sequences) - 1
            idx, prio, weight = self._ext.sumtree_sample(
                self.tree, self.leaf_offset, self.num_levels, jitter, B,
                self.beta, max_leaf)
            meta, seg = self._ext.replay_gather_meta(
                idx, self.burn_s, self.learn_s, self.fwd_s, self.obs_start_s,
                self.learn_off_s, self.spb)
            outs = self._ext.replay_gather_batch(
                self.obs_store, self.la_store, self.lr_store, self.act_store,
                self.nsr_store, self.gam_store, self.hid_store, idx, meta,
                seg, weight, self.T, self.A, self.learn_len, self.H, self.spb)
            if B == self.batch_size:
                self._pin_i ^= 1
                meta_h = self._pin_meta[self._pin_i]
                seg_h = self._pin_seg[self._pin_i]
                meta_h.copy_(meta[:3], non_blocking=True)
                seg_h.copy_(seg, non_blocking=True)
            else:
                meta_h = meta[:3].cpu()
                seg_h = seg.cpu()
            self._ev_sampled.record(self._sample_stream)
        return (idx, outs, meta, seg, meta_h, seg_h, self.block_ptr,
                self.env_steps)

    def sample_wait(self, token):
        """Complete a sample_async(): host-syncs the side stream only."""
        from ..worker import TrainingBatch

        idx, outs, meta_dev, seg_dev, meta_h, seg_h, old_ptr, env_steps = token
        self._ev_sampled.synchronize()
        obs, la, lr, act, nsr, gam, w_rep, hid = outs
        B = idx.shape[0]
        R = int(seg_h[-1])
        c_, h_, w_ = self.obs_shape
        batch = TrainingBatch(
            obs=obs.view(B, self.T, h_, w_, c_),   # HWC layout
            last_action=la, last_reward=lr, hidden=hid,
            action=act[:R].unsqueeze(1), n_step_reward=nsr[:R], gamma=gam[:R],
            burn_in_steps=meta_h[0].long(), learning_steps=meta_h[1].long(),
            forward_steps=meta_h[2].long(),
            idxes=idx, is_weights=w_rep[:R],
            old_ptr=old_ptr, env_steps=env_steps)
        # device-side layout metadata: the engine computes its gather/
        # scatter position arrays from these with one kernel instead of a
        # per-batch host numpy rebuild (every replay batch is ragged)
        batch.meta_dev = meta_dev
        batch.seg_dev = seg_dev
        # training-stream consumers must see the gathered tensors
        torch.cuda.current_stream().wait_event(self._ev_sampled)
        return batch

    def sample(self, batch_size: Optional[int] = None):
        """Synchronous sampling (sample_async + sample_wait)."""
        return self.sample_wait(self.sample_async(batch_size))

    # ------------------------------------------------------------------
    def update_priorities(self, idxes: torch.Tensor, priorities: torch.Tensor,
                          old_ptr: int):
        """idxes/priorities stay on device; ring-stale mask applied in-kernel.
        Waits on any in-flight sample's tree reads (side stream).

        The host ReplayBuffer's full-ring-lap hole (see worker.py
        update_priorities: a pointer-only mask cannot distinguish "nothing
        overwritten" from "everything overwritten") does not arise here:
        priority staleness is bounded at ONE batch by construction — the
        learner updates synchronously after each step and prefetch depth
        is 1 — so the ring can never advance num_blocks slots between a
        sample and its update."""
        torch.cuda.current_stream().wait_event(self._ev_sampled)
        self._ext.sumtree_update(self.tree, self.leaf_offset,
                                 idxes.to(torch.int64),
                                 priorities.float(), self.alpha,
                                 old_ptr, self.block_ptr, self.spb,
                                 self.num_blocks)
