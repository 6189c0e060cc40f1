"""Actor / Learner / ReplayBuffer — the core R2D2 runtime.

API parity with the reference's worker.py (/root/reference/worker.py) at the
class level: ``Block``, ``LocalBuffer``, ``ReplayBuffer``, ``Learner``,
``Actor``, ``calculate_mixed_td_errors``, ``Learner.value_rescale``.  The
implementation is new and MI355X-first:

- the learner runs ONE online forward per update (``calculate_q_both``)
  instead of the reference's two (worker.py:346,352), in bf16 on GPU;
- batch assembly is vectorized numpy (no per-sample Python slice loop as in
  reference worker.py:176-210);
- the replay ingest thread blocks on a queue instead of busy-spinning
  (reference worker.py:124-129 burns a core);
- with ``config.gpu_replay`` the prioritized replay lives in GPU HBM with
  on-device sampling/priority updates (replay/gpu_replay.py) and priorities
  never leave the device;
- stored recurrent states are aligned to each sequence's burn-in start
  (the reference stores index ``i*learning_steps`` into the carried buffer,
  which is only correct once the burn-in prefix is full — reference
  worker.py:461 vs :186-188).

Checkpoint format is the reference 4-tuple
``(state_dict, num_updates, env_steps, minutes)`` (worker.py:380-381).
"""

import math
import os
import queue as queue_mod
import threading
import time
from dataclasses import dataclass
from typing import List, Optional

import numpy as np
import torch
import torch.nn as nn

from . import config as cfg
from .models.network import Network, AgentState
from .ops import functional as Fn
from .replay.priority_tree import PriorityTree


############################## Data containers ##############################


@dataclass
class Block:
    """Unit of actor->replay transfer: <=block_length transitions cut into
    <=seq_per_block sequences of learning_steps each (reference worker.py:23-36)."""
    obs: np.ndarray            # (burn_in0 + steps + 1, *obs_shape) uint8/float32
    last_action: np.ndarray    # (burn_in0 + steps + 1, A) bool one-hot
    last_reward: np.ndarray    # (burn_in0 + steps + 1,) float32
    action: np.ndarray         # (steps,) uint8
    n_step_reward: np.ndarray  # (steps,) float32
    gamma: np.ndarray          # (steps,) float32 — per-step gamma^n, 0 at terminal
    hidden: np.ndarray         # (num_sequences, 2, hidden_dim) float32
    num_sequences: int
    burn_in_steps: np.ndarray  # (num_sequences,) uint8
    learning_steps: np.ndarray # (num_sequences,) uint8
    forward_steps: np.ndarray  # (num_sequences,) uint8


@dataclass
class TrainingBatch:
    """Device-ready training batch (the reference ships an anonymous 14-tuple,
    worker.py:219-238; named fields here, same content)."""
    obs: torch.Tensor            # (B, T, *obs_shape)
    last_action: torch.Tensor    # (B, T, A)
    last_reward: torch.Tensor    # (B, T)
    hidden: torch.Tensor         # (2, B, H)
    action: torch.Tensor         # (sum_learn, 1) long
    n_step_reward: torch.Tensor  # (sum_learn,)
    gamma: torch.Tensor          # (sum_learn,)
    burn_in_steps: torch.Tensor  # (B,) CPU
    learning_steps: torch.Tensor # (B,) CPU
    forward_steps: torch.Tensor  # (B,) CPU
    idxes: np.ndarray
    is_weights: torch.Tensor     # (sum_learn,)
    old_ptr: int
    env_steps: int
    old_count: int = 0           # monotone blocks-added at sample time

    def to(self, device, non_blocking=True):
        self.obs = self.obs.to(device, non_blocking=non_blocking)
        self.last_action = self.last_action.to(device, non_blocking=non_blocking)
        self.last_reward = self.last_reward.to(device, non_blocking=non_blocking)
        self.hidden = self.hidden.to(device, non_blocking=non_blocking)
        self.action = self.action.to(device, non_blocking=non_blocking)
        self.n_step_reward = self.n_step_reward.to(device, non_blocking=non_blocking)
        self.gamma = self.gamma.to(device, non_blocking=non_blocking)
        self.is_weights = self.is_weights.to(device, non_blocking=non_blocking)
        return self

    def pin(self):
        for f in ("obs", "last_action", "last_reward", "hidden", "action",
                  "n_step_reward", "gamma", "is_weights"):
            setattr(self, f, getattr(self, f).pin_memory())
        return self


def _shared_tensor(shape, dtype) -> torch.Tensor:
    """CPU tensor allocated DIRECTLY in shared memory (no intermediate
    copy).  A torch.multiprocessing queue ships an already-shared tensor
    as a handle; a private tensor gets copied into shared memory by the
    queue's single feeder thread.  At the full MsPacman batch shape the
    obs tensor is ~150 MB, and that serialized feeder copy — not the
    assembly itself — bounded the host-replay pipeline (measured on this
    container: assembler alone 33-60 batches/s at 2-4 threads, but only
    7-10.5 batches/s through the queue; handle-passing removes the wall)."""
    t = torch.empty(0, dtype=dtype)
    numel = int(np.prod(shape))
    storage = torch.UntypedStorage._new_shared(numel * t.element_size())
    return t.set_(storage, 0, tuple(shape))


def calculate_mixed_td_errors(td_error: np.ndarray, learning_steps: np.ndarray,
                              eta: float = 0.9) -> np.ndarray:
    """Per-sequence priority = eta*max + (1-eta)*mean of |TD|
    (reference worker.py:268-276)."""
    return Fn.mixed_td_priority_np(td_error, learning_steps, eta)


############################## LocalBuffer ##############################


class LocalBuffer:
    """Per-actor trajectory accumulator (reference worker.py:395-497).

    Stores one episode's running transitions, cuts them into blocks of
    <= block_length steps, computes on-actor: n-step returns, the per-step
    gamma^n vector (0 at terminal -> no done flag stored), initial priorities
    from the actor's own q-values, and carries the last burn_in+1 steps as
    the next block's burn-in prefix.
    """

    def __init__(self, action_dim: int, forward_steps: Optional[int] = None,
                 burn_in_steps: Optional[int] = None, learning_steps: Optional[int] = None,
                 gamma: Optional[float] = None, hidden_dim: Optional[int] = None,
                 block_length: Optional[int] = None):
        c = cfg.get()
        self.action_dim = action_dim
        self.gamma = c.gamma if gamma is None else gamma
        self.hidden_dim = c.hidden_dim if hidden_dim is None else hidden_dim
        self.forward_steps = c.forward_steps if forward_steps is None else forward_steps
        self.learning_steps = c.learning_steps if learning_steps is None else learning_steps
        self.burn_in_steps = c.burn_in_steps if burn_in_steps is None else burn_in_steps
        self.block_length = c.block_length if block_length is None else block_length
        self.prio_eta = c.prio_eta
        self.curr_burn_in_steps = 0
        self.size = 0

    def __len__(self):
        return self.size

    def reset(self, init_obs: np.ndarray):
        self.obs_buffer: List[np.ndarray] = [init_obs]
        first_action = np.zeros(self.action_dim, dtype=bool)
        first_action[0] = True
        self.last_action_buffer: List[np.ndarray] = [first_action]
        self.last_reward_buffer: List[float] = [0.0]
        self.hidden_buffer: List[np.ndarray] = [
            np.zeros((2, self.hidden_dim), dtype=np.float32)]
        self.action_buffer: List[int] = []
        self.reward_buffer: List[float] = []
        self.qval_buffer: List[np.ndarray] = []
        self.curr_burn_in_steps = 0
        self.size = 0
        self.sum_reward = 0.0
        self.done = False

    def add(self, action: int, reward: float, next_obs: np.ndarray,
            q_value: np.ndarray, hidden_state: np.ndarray):
        """hidden_state: (2, H) — the LSTM state AFTER processing the current
        obs, i.e. the correct initial state for a sequence starting at
        next_obs."""
        self.action_buffer.append(action)
        self.reward_buffer.append(reward)
        self.obs_buffer.append(next_obs)
        onehot = np.zeros(self.action_dim, dtype=bool)
        onehot[action] = True
        self.last_action_buffer.append(onehot)
        self.last_reward_buffer.append(reward)
        self.hidden_buffer.append(hidden_state)
        self.qval_buffer.append(q_value.reshape(-1))
        self.sum_reward += reward
        self.size += 1

    def finish(self, last_qval: Optional[np.ndarray] = None):
        assert 0 < self.size <= self.block_length
        S, n, L = self.size, self.forward_steps, self.learning_steps
        num_sequences = math.ceil(S / L)
        mfs = min(S, n)

        # per-step bootstrap discount (reference worker.py:443-455 semantics)
        self.done = last_qval is None
        gamma_vec = Fn.gamma_vector(S, n, self.gamma, self.done)
        if self.done:
            self.qval_buffer.append(np.zeros_like(self.qval_buffer[0]))
        else:
            self.qval_buffer.append(last_qval.reshape(-1))

        obs = np.stack(self.obs_buffer)
        last_action = np.stack(self.last_action_buffer)
        last_reward = np.array(self.last_reward_buffer, dtype=np.float32)
        actions = np.array(self.action_buffer, dtype=np.uint8)

        # n-step return over this block's rewards (zero-padded tail)
        n_step_reward = Fn.n_step_return(
            np.array(self.reward_buffer, dtype=np.float32), n, self.gamma)

        # per-sequence layout
        burn_in = np.array([min(i * L + self.curr_burn_in_steps, self.burn_in_steps)
                            for i in range(num_sequences)], dtype=np.uint8)
        learning = np.array([min(L, S - i * L) for i in range(num_sequences)],
                            dtype=np.uint8)
        forward = np.array([min(n, S + 1 - int(np.sum(learning[:i + 1])))
                            for i in range(num_sequences)], dtype=np.uint8)
        assert forward[-1] == 1 and burn_in[0] == self.curr_burn_in_steps

        # stored recurrent state, aligned to each sequence's burn-in start:
        # learning start (buffer index) = curr_burn_in + i*L; burn-in start =
        # that - burn_in[i].  (The reference stores index i*L, worker.py:461,
        # which matches only when the carried prefix is full.)
        h_idx = [self.curr_burn_in_steps + i * L - int(burn_in[i])
                 for i in range(num_sequences)]
        hiddens = np.stack([self.hidden_buffer[j] for j in h_idx])

        # initial priorities from the actor's own q-values
        qvals = np.stack(self.qval_buffer)          # (S+1, A); current block steps
        max_q = qvals[mfs: S + 1].max(axis=1)
        max_q = np.pad(max_q, (0, mfs - 1), mode="edge") if mfs > 1 else max_q
        taken_q = qvals[np.arange(S), actions]
        td_errors = np.abs(n_step_reward + gamma_vec * max_q - taken_q).astype(np.float32)
        priorities = np.zeros(self.block_length // L, dtype=np.float32)
        priorities[:num_sequences] = calculate_mixed_td_errors(
            td_errors, learning, self.prio_eta)

        block = Block(obs, last_action, last_reward, actions, n_step_reward,
                      gamma_vec, hiddens, num_sequences, burn_in, learning, forward)

        # carry burn-in prefix into the next block
        keep = self.burn_in_steps + 1
        self.obs_buffer = self.obs_buffer[-keep:]
        self.last_action_buffer = self.last_action_buffer[-keep:]
        self.last_reward_buffer = self.last_reward_buffer[-keep:]
        self.hidden_buffer = self.hidden_buffer[-keep:]
        self.action_buffer.clear()
        self.reward_buffer.clear()
        self.qval_buffer.clear()
        self.curr_burn_in_steps = len(self.obs_buffer) - 1
        self.size = 0

        return [block, priorities, self.sum_reward if self.done else None]


############################## ReplayBuffer ##############################


class ReplayBuffer:
    """Prioritized block ring buffer + batch assembler (host mode).

    Runs in its own process (reference worker.py:38-261) with three worker
    threads: ingest (actor queues -> ring), assemble (tree sample -> batches
    -> batch_queue), and priority-update.  Also the console logger and the
    run-termination condition.
    """

    def __init__(self, sample_queue_list, batch_queue, priority_queue,
                 buffer_capacity: Optional[int] = None,
                 alpha: Optional[float] = None, beta: Optional[float] = None,
                 batch_size: Optional[int] = None, seed: Optional[int] = None,
                 metrics_path: Optional[str] = None,
                 restore_path: Optional[str] = None,
                 initial_training_steps: int = 0):
        c = cfg.get()
        self.cfg = c
        self.block_len = c.block_length
        self.seq_len = c.learning_steps
        self.buffer_capacity = buffer_capacity or c.buffer_capacity
        self.num_sequences = self.buffer_capacity // self.seq_len
        self.num_blocks = self.buffer_capacity // self.block_len
        self.seq_per_block = self.block_len // self.seq_len
        self.batch_size = batch_size or c.batch_size

        self.priority_tree = PriorityTree(
            self.num_sequences,
            alpha if alpha is not None else c.prio_exponent,
            beta if beta is not None else c.importance_sampling_exponent,
            rng=np.random.default_rng(seed))

        self.block_ptr = 0
        self.blocks_added = 0    # monotone (never wraps)
        self.size = 0
        self.env_steps = 0
        self.num_episodes = 0
        self.episode_reward = 0.0
        # on resume the learner continues from a restored update count; the
        # buffer's run-termination condition must count from the same origin
        self.training_steps = initial_training_steps
        self.last_training_steps = initial_training_steps
        self.sum_loss = 0.0
        self.last_size = 0
        self.lock = threading.Lock()
        self.buffer: List[Optional[Block]] = [None] * self.num_blocks
        self.sample_queue_list = sample_queue_list
        self.batch_queue = batch_queue
        self.priority_queue = priority_queue
        self.stop_flag = False
        # observability: each _log() tick also appends one JSONL record
        # (SURVEY §5 — the reference only print()s, worker.py:89-111)
        self.metrics_path = metrics_path or getattr(c, "metrics_path", None)
        self._t0 = time.time()
        # rotating pool of shared-memory obs tensors for outgoing batches
        # (see _shared_tensor / sample_batch).  Safety contract: a slot is
        # reused after pool-depth claims, so no consumer may still hold a
        # batch once pool-depth NEWER batches were claimed.  The topology
        # enforces that by throttling: assemblers only produce while
        # batch_queue has room, so claims past the oldest live batch are
        # bounded by queue depth + learner staging (<=5) + one per
        # assemble thread — the pool is sized at 2x that bound.
        n_slots = max(32, 2 * (int(getattr(c, "batch_queue_size", 8))
                               + int(getattr(c, "assemble_threads", 2)) + 6))
        self._obs_pool: List[Optional[torch.Tensor]] = [None] * n_slots
        self._obs_pool_i = 0
        # replay persistence (elastic resume): snapshot cadence handled in
        # run(); restore happens before the worker threads start
        self.snapshot_path = getattr(c, "replay_snapshot_path", None)
        self.snapshot_interval = getattr(c, "replay_snapshot_interval", 300.0)
        if restore_path:
            self.load_state(restore_path)

    def __len__(self):
        return self.size

    # -- threads -----------------------------------------------------------

    def run(self):
        # Complete torch's lazy parallel-backend init on THIS thread before
        # the assemble threads start: in a forked child (this process) two
        # threads racing the first parallel-sized torch op can hit a dead
        # inherited OpenMP pool ("Invalid thread pool!", ParallelOpenMP.cpp
        # lazy-init race) and kill an assembler at startup.  Only reachable
        # at full shapes — the op must cross the parallel grain size.
        torch.set_num_threads(1)
        torch.zeros((64, 2, 512)).transpose(0, 1).contiguous()
        n_asm = max(1, int(getattr(self.cfg, "assemble_threads", 2)))
        threads = [threading.Thread(target=f, daemon=True)
                   for f in ((self._ingest_loop, self._priority_loop)
                             + (self._assemble_loop,) * n_asm)]
        for t in threads:
            t.start()
        log_interval = self.cfg.log_interval
        last_snapshot = time.time()
        while True:
            self._log(log_interval)
            if (self.snapshot_path and self.size > 0
                    and time.time() - last_snapshot >= self.snapshot_interval):
                try:
                    self.save_state(self.snapshot_path)
                except OSError as e:
                    print(f"[buffer] replay snapshot failed: {e!r}")
                last_snapshot = time.time()
            if self.training_steps >= self.cfg.training_steps:
                self.stop_flag = True
                break
            time.sleep(log_interval)

    def _log(self, interval):
        rec = {
            "wall_s": round(time.time() - self._t0, 1),
            "buffer_size": self.size,
            "buffer_fill_rate": (self.size - self.last_size) / interval,
            "env_steps": self.env_steps,
            "training_steps": self.training_steps,
        }
        print(f"buffer size: {self.size}")
        print(f"buffer update speed: {rec['buffer_fill_rate']}/s")
        self.last_size = self.size
        print(f"number of environment steps: {self.env_steps}")
        if self.num_episodes:
            rec["avg_episode_return"] = self.episode_reward / self.num_episodes
            print(f"average episode return: {rec['avg_episode_return']:.4f}")
            self.episode_reward = 0.0
            self.num_episodes = 0
        print(f"number of training steps: {self.training_steps}")
        delta = self.training_steps - self.last_training_steps
        rec["training_rate"] = delta / interval
        rec["seq_samples_per_s"] = rec["training_rate"] * self.batch_size
        print(f"training speed: {delta / interval}/s")
        if delta:
            rec["loss"] = self.sum_loss / delta
            print(f"loss: {rec['loss']:.4f}")
            self.last_training_steps = self.training_steps
            self.sum_loss = 0.0
        print()
        if self.metrics_path:
            import json
            with open(self.metrics_path, "a") as f:
                f.write(json.dumps(rec) + "\n")

    def _ingest_loop(self):
        """Blocking multiplexed ingest (no busy-spin)."""
        while not self.stop_flag:
            got = False
            for q in self.sample_queue_list:
                try:
                    data = q.get_nowait()
                except queue_mod.Empty:
                    continue
                self.add(*data)
                got = True
            if not got:
                time.sleep(0.005)

    def _assemble_loop(self):
        while self.size < self.cfg.learning_starts and not self.stop_flag:
            time.sleep(0.5)
        while not self.stop_flag:
            if not self.batch_queue.full():
                self.batch_queue.put(self.sample_batch())
            else:
                time.sleep(0.02)

    def _priority_loop(self):
        while not self.stop_flag:
            try:
                data = self.priority_queue.get(timeout=0.1)
            except queue_mod.Empty:
                continue
            self.update_priorities(*data)

    # -- operations ---------------------------------------------------------

    def add(self, block: Block, priority: np.ndarray, episode_reward):
        with self.lock:
            idxes = np.arange(self.block_ptr * self.seq_per_block,
                              (self.block_ptr + 1) * self.seq_per_block,
                              dtype=np.int64)
            self.priority_tree.update(idxes, priority)
            old = self.buffer[self.block_ptr]
            if old is not None:
                self.size -= int(np.sum(old.learning_steps))
            self.size += int(np.sum(block.learning_steps))
            self.env_steps += int(np.sum(block.learning_steps))
            self.buffer[self.block_ptr] = block
            self.block_ptr = (self.block_ptr + 1) % self.num_blocks
            self.blocks_added += 1
            if episode_reward is not None:
                self.episode_reward += episode_reward
                self.num_episodes += 1

    def sample_batch(self) -> TrainingBatch:
        """Vectorized batch assembly (replaces the reference's per-sample
        Python slice loop, worker.py:176-210).

        The LOCK covers only the tree sample and capturing Block
        REFERENCES (a Block is immutable once stored; a ring overwrite
        replaces the list slot but our reference keeps the sampled data
        alive and coherent).  The heavy slice copies run unlocked, so
        several assemble threads overlap (the copies are memcpys that
        release the GIL) and ingest/priority threads never stall behind
        an assembly."""
        with self.lock:
            idxes, is_weights = self.priority_tree.sample(self.batch_size)
            block_idxes = idxes // self.seq_per_block
            seq_idxes = idxes % self.seq_per_block
            blocks = []
            for bi, si in zip(block_idxes, seq_idxes):
                blk = self.buffer[bi]
                assert blk is not None and si < blk.num_sequences
                blocks.append(blk)
            old_ptr = self.block_ptr
            old_count = self.blocks_added
            env_steps = self.env_steps
            # claim the next shared-memory obs slot while the lock is held
            pool_i = self._obs_pool_i
            self._obs_pool_i = (pool_i + 1) % len(self._obs_pool)

        B = self.batch_size
        burn = np.empty(B, dtype=np.int64)
        learn = np.empty(B, dtype=np.int64)
        fwd = np.empty(B, dtype=np.int64)
        starts = np.empty(B, dtype=np.int64)
        for i, (blk, si) in enumerate(zip(blocks, seq_idxes)):
            burn[i] = blk.burn_in_steps[si]
            learn[i] = blk.learning_steps[si]
            fwd[i] = blk.forward_steps[si]
            # int() both operands: a uint8 SCALAR + a >255 python int
            # raises OverflowError under NumPy 2 promotion (NEP 50) —
            # reachable only at full block scale (learn sums > 255)
            starts[i] = (int(blk.burn_in_steps[0])
                         + int(np.sum(blk.learning_steps[:si])))

        if True:
            T = int((burn + learn + fwd).max())
            obs_shape = blocks[0].obs.shape[1:]
            A = blocks[0].last_action.shape[1]
            # the big obs tensor is built in shared memory so the
            # batch_queue passes a handle instead of feeder-copying
            # ~150 MB per batch (see _shared_tensor).  Slots are pooled:
            # a FRESH shm segment would cold-page-fault its whole extent
            # on every fill (measured: halves the assembler and caps it
            # ~17 batches/s regardless of threads); reused slots stay
            # warm.  Allocated at the max sequence length, sliced to this
            # batch's T.
            dt = torch.from_numpy(blocks[0].obs[:1]).dtype
            Tmax = max(T, self.cfg.seq_len)
            slot = self._obs_pool[pool_i]
            if (slot is None or slot.dtype != dt
                    or slot.shape[1] < Tmax or slot.shape[2:] != obs_shape):
                slot = _shared_tensor((B, Tmax) + obs_shape, dt)
                self._obs_pool[pool_i] = slot
            obs_t = slot[:, :T]
            obs = obs_t.numpy()   # rows are fully written below; only the
                                  # ragged padding tail needs zeroing
            last_action = np.zeros((B, T, A), dtype=np.float32)
            last_reward = np.zeros((B, T), dtype=np.float32)
            hidden = np.empty((B, 2, blocks[0].hidden.shape[-1]), dtype=np.float32)
            actions, rewards, gammas = [], [], []
            for i, blk in enumerate(blocks):
                si = seq_idxes[i]
                s, L = starts[i], int(burn[i] + learn[i] + fwd[i])
                obs[i, :L] = blk.obs[s - burn[i]: s + learn[i] + fwd[i]]
                obs[i, L:] = 0   # padding tail (slot is reused, not fresh)
                last_action[i, :L] = blk.last_action[s - burn[i]: s + learn[i] + fwd[i]]
                last_reward[i, :L] = blk.last_reward[s - burn[i]: s + learn[i] + fwd[i]]
                hidden[i] = blk.hidden[si]
                ls = int(np.sum(blk.learning_steps[:si]))
                le = ls + int(learn[i])
                actions.append(blk.action[ls:le])
                rewards.append(blk.n_step_reward[ls:le])
                gammas.append(blk.gamma[ls:le])

            is_rep = np.repeat(is_weights, learn).astype(np.float32)
            batch = TrainingBatch(
                obs=obs_t,
                last_action=torch.from_numpy(last_action),
                last_reward=torch.from_numpy(last_reward),
                hidden=torch.from_numpy(hidden).transpose(0, 1).contiguous(),
                action=torch.from_numpy(np.concatenate(actions)).long().unsqueeze(1),
                n_step_reward=torch.from_numpy(np.concatenate(rewards)),
                gamma=torch.from_numpy(np.concatenate(gammas)),
                burn_in_steps=torch.from_numpy(burn),
                learning_steps=torch.from_numpy(learn),
                forward_steps=torch.from_numpy(fwd),
                idxes=idxes,
                is_weights=torch.from_numpy(is_rep),
                old_ptr=old_ptr,
                env_steps=env_steps,
                old_count=old_count,
            )
        return batch

    def update_priorities(self, idxes: np.ndarray, td_errors: np.ndarray,
                          old_ptr: int, loss: float,
                          old_count: Optional[int] = None):
        """Masks out indexes overwritten by the ring pointer since sampling
        (reference worker.py:242-261 wraparound semantics), then tree update.

        The pointer comparison alone CANNOT see a full ring lap: if the
        ring advanced >= num_blocks blocks between sampling and this
        (delayed) update, cur == old looks like "nothing overwritten" and
        the stale priorities resurrect slots of newer, possibly SHORTER
        blocks — dead sequence slots with nonzero priority then crash the
        assembler.  ``old_count`` (monotone blocks-added at sample time)
        closes the hole; the reference shares this flaw."""
        with self.lock:
            if (old_count is not None
                    and self.blocks_added - old_count >= self.num_blocks):
                idxes = idxes[:0]
                td_errors = td_errors[:0]
            elif self.block_ptr > old_ptr:
                mask = ((idxes < old_ptr * self.seq_per_block)
                        | (idxes >= self.block_ptr * self.seq_per_block))
                idxes, td_errors = idxes[mask], td_errors[mask]
            elif self.block_ptr < old_ptr:
                mask = ((idxes < old_ptr * self.seq_per_block)
                        & (idxes >= self.block_ptr * self.seq_per_block))
                idxes, td_errors = idxes[mask], td_errors[mask]
            if len(idxes):
                self.priority_tree.update(idxes, td_errors)
        self.training_steps += 1
        self.sum_loss += loss

    # -- persistence (elastic resume) ---------------------------------------

    def save_state(self, path: str):
        """Snapshot the replay contents for elastic resume — the reference
        never persists replay (a restart refills 2M transitions
        from scratch).  Blocks are immutable once stored, so the lock only
        covers capturing the slot references and counters; serialization
        runs unlocked, and the write is atomic (tmp + rename) so a crash
        mid-snapshot leaves the previous snapshot intact."""
        import pickle

        with self.lock:
            state = {
                "geometry": (self.num_blocks, self.seq_per_block,
                             self.block_len, self.seq_len),
                "blocks": list(self.buffer),
                "leaves": self.priority_tree.leaf_values(),
                "block_ptr": self.block_ptr,
                "blocks_added": self.blocks_added,
                "size": self.size,
                "env_steps": self.env_steps,
            }
        tmp = path + ".tmp"
        with open(tmp, "wb") as f:
            pickle.dump(state, f, protocol=pickle.HIGHEST_PROTOCOL)
        os.replace(tmp, path)

    def load_state(self, path: str):
        """Restore a save_state snapshot (geometry must match the live
        config).  Call before run() — not thread-safe against the worker
        threads."""
        import pickle

        with open(path, "rb") as f:
            state = pickle.load(f)
        geo = (self.num_blocks, self.seq_per_block, self.block_len,
               self.seq_len)
        if tuple(state["geometry"]) != geo:
            raise ValueError(f"replay snapshot geometry {state['geometry']} "
                             f"!= configured {geo}")
        with self.lock:
            self.buffer = list(state["blocks"])
            self.priority_tree.set_leaf_values(state["leaves"])
            self.block_ptr = int(state["block_ptr"])
            self.blocks_added = int(state["blocks_added"])
            self.size = int(state["size"])
            self.env_steps = int(state["env_steps"])


############################## Learner ##############################


class Learner:
    """GPU learner: online + target nets, fused single-pass double-Q update,
    Adam with grad-norm clip, priority feedback, weight publish, checkpoints
    (reference worker.py:278-390)."""

    def __init__(self, batch_queue, priority_queue, model: Network,
                 grad_norm: Optional[float] = None, lr: Optional[float] = None,
                 eps: Optional[float] = None, game_name: Optional[str] = None,
                 target_net_update_interval: Optional[int] = None,
                 save_interval: Optional[int] = None,
                 model_dir: str = "models"):
        c = cfg.get()
        self.cfg = c
        self.device = torch.device(
            c.device if torch.cuda.is_available() or c.device == "cpu" else "cpu")
        self.online_net = Network(model.action_dim, model.obs_shape,
                                  model.hidden_dim, encoder=c.encoder,
                                  forward_steps=c.forward_steps,
                                  mlp_hidden=c.mlp_hidden)
        self.online_net.load_state_dict(model.state_dict())
        self.online_net.to(self.device)
        self.online_net.train()
        self.target_net = Network(model.action_dim, model.obs_shape,
                                  model.hidden_dim, encoder=c.encoder,
                                  forward_steps=c.forward_steps,
                                  mlp_hidden=c.mlp_hidden)
        self.target_net.load_state_dict(model.state_dict())
        self.target_net.to(self.device)
        self.target_net.eval()
        self.optimizer = torch.optim.Adam(self.online_net.parameters(),
                                          lr=lr or c.lr, eps=eps or c.eps)
        self.grad_norm = grad_norm or c.grad_norm
        self.batch_queue = batch_queue
        self.priority_queue = priority_queue
        self.num_updates = 0
        self.env_steps = 0
        self.target_net_update_interval = (target_net_update_interval
                                           or c.target_net_update_interval)
        self.save_interval = save_interval or c.save_interval
        self.game_name = game_name or c.game_name
        self.model_dir = model_dir
        self.shared_model = model
        self.batched_data: List[TrainingBatch] = []
        self.amp = c.amp and self.device.type == "cuda" and c.dtype == "bf16"

        # multi-learner data parallelism: bucketed grad all-reduce over
        # RCCL/xGMI, overlapped with backward (parallel/ddp.py)
        self.reducer = None
        import torch.distributed as dist
        if dist.is_initialized() and dist.get_world_size() > 1:
            from .parallel.ddp import GradAllReducer
            self.reducer = GradAllReducer(list(self.online_net.parameters()))
            self.reducer.broadcast_params(src=0)
            self.target_net.load_state_dict(self.online_net.state_dict())

    # -- HIP engine ---------------------------------------------------------

    def enable_hip_engine(self):
        """Switch to the gfx950 HIP kernels.  For the flagship nature/512
        config the FULL forward/backward runs through the hand-written engine
        (ops/engine.py); otherwise the fused loss/priority kernels are used
        under the eager forward.  Raises if the extension is not built — on a
        GPU box the native path must be the one that runs."""
        from .ops import hip_ops
        hip_ops.ext(required=True)
        self.hip_engine = True
        c = self.cfg
        if (c.encoder in ("nature", "impala") and c.hidden_dim == 512
                and len(c.obs_shape) == 3 and c.obs_shape[0] == 4
                and self.device.type == "cuda"):
            from .ops.engine import HipNetworkEngine
            self.engine = HipNetworkEngine(self.online_net, self.target_net,
                                           self.device, c)
            # device-resident weight distribution to same-GPU VectorActors
            # (replaces the reference's GPU->CPU shared-memory
            # publish + per-pull repack, worker.py:306-307,560-566).  The
            # spec comes from the target pack — same tensor names/shapes as
            # the actor-side inference pack (with_bwd=False).
            from .parallel.weight_bus import WeightBus, pack_tensors
            self.weight_bus = WeightBus(pack_tensors(self.engine.target),
                                        self.device)

    hip_engine = False
    engine = None
    weight_bus = None

    # -- weight publication -------------------------------------------------

    def store_weights(self, cpu: bool = True):
        """Publish weights: device bus (VectorActor consumers, cheap D2D)
        and/or the shared CPU model (reference worker.py:306-307 contract,
        process-Actor consumers + watchdog restarts).  When the bus serves
        the actors, the CPU publish is throttled by the caller — the 17 MB
        GPU->CPU state_dict round trip otherwise blocks the learner thread
        every 4 updates for consumers that rarely read it."""
        if cpu:
            state = {k: v.cpu()
                     for k, v in self.online_net.state_dict().items()}
            self.shared_model.load_state_dict(state)
        if self.weight_bus is not None and self.engine is not None:
            self.weight_bus.publish(self.engine.online)

    # -- data staging -------------------------------------------------------

    def _prefetch_loop(self):
        while True:
            if len(self.batched_data) < 4:
                try:
                    data = self.batch_queue.get(timeout=0.1)
                except queue_mod.Empty:
                    continue
                if self.device.type == "cuda":
                    data.pin()
                self.batched_data.append(data)
            else:
                time.sleep(0.01)

    # -- the update step ----------------------------------------------------

    def train_step(self, batch: TrainingBatch):
        """One update.  Returns (loss tensor, per-sequence priorities — a
        device tensor on the HIP path, numpy on the eager path)."""
        c = self.cfg
        batch.to(self.device)

        if self.engine is not None:
            # full HIP path: manual backward fills the flat grad buffer.
            # DP: each flat-grad segment all-reduces over RCCL as soon as
            # the manual backward completes it (heads -> lstm -> encoder),
            # overlapping the collective with the remaining backward; the
            # works are drained before the fused clip+Adam step.
            if self.reducer is not None:
                import torch.distributed as dist
                works = []
                ws = self.reducer.world_size

                def _seg_reduce(seg):
                    lo, hi = self.engine.seg_ranges[seg]
                    g = self.engine.flat_grad[lo:hi]
                    g.div_(ws)
                    works.append(dist.all_reduce(g, async_op=True))

                loss, prio = self.engine.train_step(batch,
                                                    grad_hook=_seg_reduce)
                for w in works:
                    w.wait()
            else:
                loss, prio = self.engine.train_step(batch)
            self.engine.optimizer_step(
                lr=self.optimizer.param_groups[0]["lr"],
                eps=self.optimizer.param_groups[0]["eps"],
                max_norm=self.grad_norm)
            self.engine.refresh_online()
            self.num_updates += 1
            return loss, prio

        h0 = (batch.hidden[:1], batch.hidden[1:])

        ctx = torch.autocast("cuda", dtype=torch.bfloat16) if self.amp else _nullctx()
        with ctx:
            with torch.no_grad():
                q_tgt_all = self.target_net.calculate_q_(
                    batch.obs, batch.last_action, batch.last_reward, h0,
                    batch.burn_in_steps, batch.learning_steps, batch.forward_steps)
            q_learn, q_online_tgt = self.online_net.calculate_q_both(
                batch.obs, batch.last_action, batch.last_reward, h0,
                batch.burn_in_steps, batch.learning_steps, batch.forward_steps)

        q_learn = q_learn.float()
        if self.hip_engine:
            # fused double-Q target + TD + loss + priority, one kernel pair,
            # dLoss/dQ computed in the same sweep (ops/hip/loss_kernels.hip)
            from .ops import hip_ops
            loss, prio_dev = hip_ops.fused_double_q_loss(
                q_learn, q_online_tgt.detach().float(), q_tgt_all.float(),
                batch.action, batch.n_step_reward, batch.gamma,
                batch.is_weights, batch.learning_steps,
                eps=c.rescale_eps, kappa=c.huber_kappa, loss_kind=c.loss_fn,
                eta=c.prio_eta)
        else:
            with torch.no_grad():
                target_q = Fn.double_q_target(
                    q_online_tgt.detach().float(), q_tgt_all.float(),
                    batch.n_step_reward, batch.gamma, c.rescale_eps)
            batch_q = q_learn.gather(1, batch.action).squeeze(1)
            loss = (batch.is_weights * Fn.per_step_loss(
                batch_q, target_q, c.loss_fn, c.huber_kappa)).mean()

        self.optimizer.zero_grad(set_to_none=True)
        if self.reducer is not None:
            self.reducer.prepare()
        loss.backward()
        if self.reducer is not None:
            self.reducer.finish()
        nn.utils.clip_grad_norm_(self.online_net.parameters(), self.grad_norm)
        self.optimizer.step()
        self.num_updates += 1

        if self.hip_engine:
            priorities = prio_dev  # device tensor; stays on GPU for gpu_replay
        else:
            td = (target_q - batch_q).detach().abs().cpu().numpy()
            priorities = calculate_mixed_td_errors(
                td, batch.learning_steps.numpy(), c.prio_eta)
        return loss.detach(), priorities

    def run(self):
        threading.Thread(target=self._prefetch_loop, daemon=True).start()
        start_time = time.time() - self._resumed_minutes * 60.0
        os.makedirs(self.model_dir, exist_ok=True)
        while self.num_updates < self.cfg.training_steps:
            while not self.batched_data:
                time.sleep(0.05)
            batch = self.batched_data.pop(0)
            loss, priorities = self.train_step(batch)
            loss = float(loss)
            if torch.is_tensor(priorities):
                priorities = priorities.cpu().numpy()
            self.priority_queue.put((batch.idxes, priorities, batch.old_ptr, loss,
                                     batch.old_count))
            self.env_steps = batch.env_steps
            if self.num_updates % 4 == 0:
                self.store_weights()
            if self.num_updates % self.target_net_update_interval == 0:
                self.target_net.load_state_dict(self.online_net.state_dict())
                if self.engine is not None:
                    self.engine.refresh_target()
            if self.num_updates % self.save_interval == 0:
                self.save(start_time)

    def run_with_gpu_replay(self, sample_queue_list):
        """The configs[2]/[3] topology: the learner process OWNS the
        GPU-resident prioritized replay (replay/gpu_replay.py).  Actor
        blocks arrive on mp.Queues and are ingested into the HBM block
        store by a background thread (pinned staging, copy stream);
        sampling, batch assembly, and priority updates never leave the
        device.  The separate host ReplayBuffer process (reference
        worker.py:77-138) disappears — this method is also the logger and
        the termination condition."""
        from .replay.gpu_replay import GpuReplayBuffer

        c = self.cfg
        replay = GpuReplayBuffer(device=self.device)
        stats = {"episode_reward": 0.0, "num_episodes": 0, "sum_loss": 0.0,
                 "last_updates": 0, "last_size": 0}
        lock = threading.Lock()
        stop = threading.Event()

        def _ingest():
            while not stop.is_set():
                got = False
                for q in sample_queue_list:
                    try:
                        block, prio, reward = q.get_nowait()
                    except queue_mod.Empty:
                        continue
                    try:
                        with lock:
                            replay.ingest(block, prio)
                            if reward is not None:
                                stats["episode_reward"] += reward
                                stats["num_episodes"] += 1
                    except Exception as e:   # keep the feed alive; surface it
                        print(f"[learner] replay ingest error: {e!r}")
                    got = True
                if not got:
                    time.sleep(0.005)

        threading.Thread(target=_ingest, daemon=True).start()
        start_time = time.time() - self._resumed_minutes * 60.0
        os.makedirs(self.model_dir, exist_ok=True)
        last_log = time.time()
        while len(replay) < c.learning_starts:
            time.sleep(0.2)
            if time.time() - last_log > c.log_interval:
                print(f"filling GPU replay: {len(replay)}/{c.learning_starts}")
                last_log = time.time()

        # side-stream prefetch: the NEXT batch samples/gathers on the
        # replay's sample stream while the current step trains (one step of
        # priority staleness — far inside the reference's own <=12-batch
        # staleness)
        pending = None
        # loss accumulates ON-DEVICE; the host reads it once per log
        # interval instead of syncing the stream every update
        loss_accum = (torch.zeros((), device=self.device)
                      if self.device.type == "cuda" else None)
        while self.num_updates < c.training_steps:
            with lock:
                batch = pending if pending is not None else replay.sample()
                tok = replay.sample_async()
            loss, priorities = self.train_step(batch)
            if not torch.is_tensor(priorities):   # eager path returns numpy
                priorities = torch.as_tensor(priorities, device=self.device)
            with lock:
                replay.update_priorities(batch.idxes, priorities,
                                         batch.old_ptr)
                pending = replay.sample_wait(tok)
            if loss_accum is not None:
                loss_accum += loss.detach()
            else:
                stats["sum_loss"] += float(loss)
            self.env_steps = replay.env_steps
            if self.num_updates % 4 == 0:
                # bus publish every 4 (reference cadence, cheap D2D); CPU
                # copy throttled when the bus serves the actors
                self.store_weights(cpu=(self.weight_bus is None
                                        or self.num_updates % 100 == 0))
            if self.num_updates % self.target_net_update_interval == 0:
                self.target_net.load_state_dict(self.online_net.state_dict())
                if self.engine is not None:
                    self.engine.refresh_target()
            if self.num_updates % self.save_interval == 0:
                self.save(start_time)
            now = time.time()
            if now - last_log > c.log_interval:
                interval = now - last_log
                delta = self.num_updates - stats["last_updates"]
                if loss_accum is not None:
                    stats["sum_loss"] = float(loss_accum)
                    loss_accum.zero_()
                rec = {
                    "wall_s": round(now - start_time, 1),
                    "buffer_size": len(replay),
                    "buffer_fill_rate":
                        (len(replay) - stats["last_size"]) / interval,
                    "env_steps": self.env_steps,
                    "training_steps": self.num_updates,
                    "training_rate": delta / interval,
                    "seq_samples_per_s": delta / interval * replay.batch_size,
                    "loss": stats["sum_loss"] / max(1, delta),
                }
                if stats["num_episodes"]:
                    rec["avg_episode_return"] = (stats["episode_reward"]
                                                 / stats["num_episodes"])
                    stats["episode_reward"] = 0.0
                    stats["num_episodes"] = 0
                print(f"buffer size: {rec['buffer_size']}  "
                      f"env steps: {rec['env_steps']}  "
                      f"updates: {rec['training_steps']} "
                      f"({rec['training_rate']:.1f}/s, "
                      f"{rec['seq_samples_per_s']:.0f} seq/s)  "
                      f"loss: {rec['loss']:.4f}")
                if self.cfg.metrics_path:
                    import json
                    with open(self.cfg.metrics_path, "a") as f:
                        f.write(json.dumps(rec) + "\n")
                stats["last_updates"] = self.num_updates
                stats["last_size"] = len(replay)
                stats["sum_loss"] = 0.0
                last_log = now
        stop.set()

    def save(self, start_time):
        """Write the reference 4-tuple checkpoint (worker.py:380-381 /
        test.py:27) plus a ``.train.pth`` sidecar (optimizer + target net)
        that enables exact resume — the reference has no resume path
        (SURVEY §5); the sidecar adds one without changing the 4-tuple
        contract the eval harness consumes."""
        os.makedirs(self.model_dir, exist_ok=True)
        state = {k: v.detach().cpu().clone() for k, v in
                 self.online_net.state_dict().items()}
        minutes = (time.time() - start_time) / 60
        stem = os.path.join(self.model_dir, f"{self.game_name}{self.num_updates}")
        torch.save((state, self.num_updates, self.env_steps, minutes),
                   stem + ".pth")
        extra = {
            "optimizer": self.optimizer.state_dict(),
            "target_net": {k: v.detach().cpu().clone() for k, v in
                           self.target_net.state_dict().items()},
            "num_updates": self.num_updates,
            "env_steps": self.env_steps,
            "minutes": minutes,
        }
        if self.engine is not None:
            # HIP path: Adam runs in the engine's fused flat-buffer kernels,
            # not the torch optimizer — persist those moments.
            extra["engine_adam"] = {
                "exp_avg": self.engine.exp_avg.cpu().clone(),
                "exp_avg_sq": self.engine.exp_avg_sq.cpu().clone(),
                "adam_t": self.engine.adam_t,
            }
        torch.save(extra, stem + ".train.pth")

    _resumed_minutes = 0.0

    def load_checkpoint(self, path: str):
        """Resume from a 4-tuple checkpoint; restores optimizer/target state
        from the ``.train.pth`` sidecar when present (else target := online,
        fresh optimizer)."""
        state, num_updates, env_steps, minutes = torch.load(
            path, map_location="cpu", weights_only=False)
        try:
            self.online_net.load_state_dict(state)
        except RuntimeError:
            # checkpoint trained by the ORIGINAL reference (anonymous
            # `feature` Sequential encoder keys, reference model.py:39-49)
            from .models.network import reference_state_dict_to_native
            state = reference_state_dict_to_native(state)
            self.online_net.load_state_dict(state)
        self.num_updates = int(num_updates)
        self.env_steps = int(env_steps)
        self._resumed_minutes = float(minutes)
        sidecar = path[:-len(".pth")] + ".train.pth"
        if os.path.exists(sidecar):
            extra = torch.load(sidecar, map_location="cpu", weights_only=False)
            self.optimizer.load_state_dict(extra["optimizer"])
            self.target_net.load_state_dict(extra["target_net"])
            if self.engine is not None and "engine_adam" in extra:
                ea = extra["engine_adam"]
                self.engine.exp_avg.copy_(ea["exp_avg"].to(self.device))
                self.engine.exp_avg_sq.copy_(ea["exp_avg_sq"].to(self.device))
                self.engine.adam_t = int(ea["adam_t"])
        else:
            self.target_net.load_state_dict(state)
        if self.engine is not None:
            self.engine.refresh_online()
            self.engine.refresh_target()
        self.store_weights()

    # reference-compat statics (worker.py:383-390)
    value_rescale = staticmethod(Fn.value_rescale)
    inverse_value_rescale = staticmethod(Fn.inverse_value_rescale)


class _nullctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


############################## Actor ##############################


class Actor:
    """CPU env loop: eps-greedy inference, LocalBuffer accumulation, periodic
    weight pull from the shared model (reference worker.py:500-574)."""

    def __init__(self, epsilon: float, model: Network, sample_queue,
                 env_fn=None, max_episode_steps: Optional[int] = None,
                 block_length: Optional[int] = None, seed: Optional[int] = None):
        c = cfg.get()
        from .envs import create_env
        self.env = env_fn() if env_fn else create_env(seed=seed)
        self.action_dim = self.env.action_dim
        self.model = Network(self.action_dim, c.obs_shape, c.hidden_dim,
                             encoder=c.encoder, forward_steps=c.forward_steps,
                             mlp_hidden=c.mlp_hidden)
        self.model.eval()
        self.local_buffer = LocalBuffer(self.action_dim)
        self.epsilon = epsilon
        self.shared_model = model
        self.sample_queue = sample_queue
        self.max_episode_steps = max_episode_steps or c.max_episode_steps
        self.block_length = block_length or c.block_length
        self.update_interval = c.actor_update_interval
        self.rng = np.random.default_rng(seed)
        self.stop_after_steps = None  # test hook

    def run(self):
        actor_steps = 0
        while True:
            done = False
            agent_state = self.reset()
            episode_steps = 0
            while not done and episode_steps < self.max_episode_steps:
                with torch.no_grad():
                    q_value, hidden = self.model(agent_state)
                if self.rng.random() < self.epsilon:
                    action = int(self.rng.integers(self.action_dim))
                else:
                    action = int(torch.argmax(q_value, 1).item())

                next_obs, reward, done, _ = self.env.step(action)
                agent_state.update(next_obs[None], action, [reward], hidden)
                episode_steps += 1
                actor_steps += 1
                self.local_buffer.add(action, reward, next_obs,
                                      q_value.numpy(),
                                      torch.cat(hidden).squeeze(1).numpy())

                if done:
                    self.sample_queue.put(self.local_buffer.finish())
                elif (len(self.local_buffer) == self.block_length
                      or episode_steps == self.max_episode_steps):
                    with torch.no_grad():
                        q_value, _ = self.model(agent_state)
                    data = self.local_buffer.finish(q_value.numpy())
                    if self.epsilon > 0.01:
                        data[2] = None  # only near-greedy actors report returns
                    self.sample_queue.put(data)

                if actor_steps % self.update_interval == 0:
                    self.update_weights()
                if self.stop_after_steps and actor_steps >= self.stop_after_steps:
                    return actor_steps

    def update_weights(self):
        self.model.load_state_dict(self.shared_model.state_dict())

    def reset(self):
        obs = self.env.reset()
        self.local_buffer.reset(obs)
        return AgentState(torch.from_numpy(obs).unsqueeze(0), self.action_dim)


############################## VectorActor ##############################


class VectorActor:
    """Batched-inference actor driver: E environments stepped in lockstep
    with ONE batched network forward per tick (SURVEY §2.4 — scaling the
    reference's 8 single-env CPU actor processes, worker.py:500-574, to
    hundreds of envs by moving inference to a single GPU context).

    Each env keeps its own LocalBuffer, epsilon (the reference ladder,
    train.py:15-17), and episode; the network forward runs once per tick
    over the (E, ...) batch — on cuda this is the K15 T=1 fast path and a
    single H2D frame copy per tick instead of E per-process forwards.

    Block cuts are finished one tick late: the reference finishes a full
    block with the q-values of the *post-step* state (worker.py:552-554);
    here that q arrives with the next tick's batched forward, so full
    buffers are flagged and finished at the top of the next tick with
    exactly the same values.
    """

    def __init__(self, epsilons: List[float], model: Network, sample_queues,
                 num_envs: Optional[int] = None, device: str = "cpu",
                 seed: Optional[int] = None, weight_bus=None):
        c = cfg.get()
        self.cfg = c
        E = num_envs or len(epsilons)
        assert len(epsilons) == E and len(sample_queues) in (1, E)
        self.E = E
        self.epsilons = np.asarray(epsilons, dtype=np.float64)
        self.device = torch.device(device if torch.cuda.is_available()
                                   or device == "cpu" else "cpu")
        from .envs import create_env
        self.envs = [create_env(seed=None if seed is None else seed + i)
                     for i in range(E)]
        self.action_dim = self.envs[0].action_dim
        self.model = Network(self.action_dim, c.obs_shape, c.hidden_dim,
                             encoder=c.encoder, forward_steps=c.forward_steps,
                             mlp_hidden=c.mlp_hidden)
        self.model.load_state_dict(model.state_dict())
        self.model.to(self.device).eval()
        self.shared_model = model
        self.queues = sample_queues
        self.buffers = [LocalBuffer(self.action_dim) for _ in range(E)]
        self.rng = np.random.default_rng(seed)
        # weight-pull cadence measured in TOTAL env-steps across the vector
        # (the reference pulls every actor_update_interval per-actor steps,
        # worker.py:560; per-tick pulling at E=256 was 1.6x too frequent and
        # paid a full load_state_dict + prepack refresh every tick)
        self.update_interval = c.actor_update_interval
        self._steps_since_pull = 0
        self.max_episode_steps = c.max_episode_steps
        self.block_length = c.block_length
        H = c.hidden_dim
        # batched device-side inference state
        self.obs_t = torch.zeros((E,) + tuple(c.obs_shape), dtype=torch.uint8,
                                 device=self.device)
        self.la_t = torch.zeros(E, self.action_dim, device=self.device)
        self.la_t[:, 0] = 1.0
        self.lr_t = torch.zeros(E, 1, device=self.device)
        self.h_t = torch.zeros(1, E, H, device=self.device)
        self.c_t = torch.zeros(1, E, H, device=self.device)
        self._obs_host = np.zeros((E,) + tuple(c.obs_shape), dtype=np.uint8)
        self.episode_steps = np.zeros(E, dtype=np.int64)
        # (env_idx, need_reset): blocks finished one tick late with the next
        # forward's bootstrap q; need_reset marks episode truncation at
        # max_episode_steps (finish, then reset like the reference's loop
        # exit at worker.py:526)
        self._pending_finish: List[tuple] = []
        # K15 fast path: batched single-step inference through the gfx950
        # kernels when running on a GPU with the extension built
        self.hip_inf = None
        self.weight_bus = weight_bus
        self._bus_ver = 0
        if (self.device.type == "cuda" and c.use_hip_kernels
                and c.encoder in ("nature", "impala") and c.hidden_dim == 512
                and tuple(c.obs_shape) == (4, 84, 84)):
            from .ops.engine import HipInference
            self.hip_inf = HipInference(self.model, self.device)
            if self.weight_bus is not None:   # catch up if already published
                self._bus_ver = self.weight_bus.pull_into(
                    self.hip_inf.pack, 0)

    def _queue(self, i):
        return self.queues[i % len(self.queues)]

    def _reset_env(self, i):
        obs = self.envs[i].reset()
        self.buffers[i].reset(obs)
        self._obs_host[i] = obs
        self.obs_t[i] = torch.from_numpy(np.ascontiguousarray(obs)).to(
            self.device, non_blocking=True)
        self.la_t[i].zero_(); self.la_t[i, 0] = 1.0
        self.lr_t[i].zero_()
        self.h_t[0, i].zero_(); self.c_t[0, i].zero_()
        self.episode_steps[i] = 0

    _started = False

    def run(self, stop_after_steps: Optional[int] = None):
        """Drive all envs until stop_after_steps total env steps (None =
        forever).  Returns total env steps taken.  Re-entrant: repeated
        calls RESUME the envs and local buffers where the previous call
        stopped (resetting would discard partially-filled blocks)."""
        if not self._started:
            for i in range(self.E):
                self._reset_env(i)
            self._started = True
        total_steps = 0
        tick = 0
        use_amp = (self.device.type == "cuda" and self.cfg.dtype == "bf16")
        state = AgentState.__new__(AgentState)  # batched raw-tensor state
        while stop_after_steps is None or total_steps < stop_after_steps:
            state.obs = self.obs_t
            state.last_action = self.la_t
            state.last_reward = self.lr_t
            state.hidden_state = (self.h_t, self.c_t)
            with torch.no_grad():
                if self.hip_inf is not None:
                    q, (h, c) = self.hip_inf.forward(
                        self.obs_t, self.la_t, self.lr_t,
                        (self.h_t, self.c_t))
                elif use_amp:
                    with torch.autocast("cuda", dtype=torch.bfloat16):
                        q, (h, c) = self.model(state)
                    q = q.float()
                else:
                    q, (h, c) = self.model(state)
            q_cpu = q.cpu().numpy()                       # (E, A)
            hid_cpu = torch.stack((h[0], c[0]), dim=1).float().cpu().numpy()

            skip = set()       # envs reset this tick: idle for one step
            for i, need_reset in self._pending_finish:    # late block cuts
                data = self.buffers[i].finish(q_cpu[i])
                if self.epsilons[i] > 0.01:
                    data[2] = None   # non-done cut: no return report
                self._queue(i).put(data)
                if need_reset:       # truncation at max_episode_steps
                    self._reset_env(i)
                    skip.add(i)
            self._pending_finish.clear()

            greedy = q_cpu.argmax(axis=1)
            explore = self.rng.random(self.E) < self.epsilons
            rand_a = self.rng.integers(0, self.action_dim, self.E)
            actions = np.where(explore, rand_a, greedy)
            if skip:
                # reset envs don't step this tick; action slot 0 matches the
                # LocalBuffer.reset() initial one-hot
                actions[list(skip)] = 0

            rewards = np.zeros(self.E, dtype=np.float32)
            dones = np.zeros(self.E, dtype=bool)
            for i, env in enumerate(self.envs):
                if i in skip:
                    continue
                a = int(actions[i])
                next_obs, r, done, _ = env.step(a)
                self.buffers[i].add(a, float(r), next_obs, q_cpu[i], hid_cpu[i])
                self._obs_host[i] = next_obs
                rewards[i] = r
                dones[i] = done
            total_steps += self.E - len(skip)
            self.episode_steps += 1
            tick += 1

            # advance batched state (one H2D copy for all frames)
            self.obs_t.copy_(torch.from_numpy(self._obs_host),
                             non_blocking=True)
            self.la_t.zero_()
            self.la_t[np.arange(self.E), actions] = 1.0
            self.lr_t.copy_(torch.from_numpy(rewards).unsqueeze(1))
            self.h_t, self.c_t = h.detach(), c.detach()
            for i in skip:   # _reset_env ran before the h/c advance: re-zero
                self.h_t[0, i].zero_()
                self.c_t[0, i].zero_()
                self.episode_steps[i] = 0

            for i in range(self.E):
                if i in skip:
                    continue
                if dones[i]:
                    # episode end: report the return from every actor
                    # (the reference filters only the non-done block cut,
                    # worker.py:555-557)
                    self._queue(i).put(self.buffers[i].finish())
                    self._reset_env(i)
                elif (len(self.buffers[i]) == self.block_length
                      or self.episode_steps[i] >= self.max_episode_steps):
                    self._pending_finish.append(
                        (i, self.episode_steps[i] >= self.max_episode_steps))

            self._steps_since_pull += self.E - len(skip)
            if self._steps_since_pull >= self.update_interval:
                self._steps_since_pull = 0
                self.pull_weights()
        return total_steps

    _pulls = 0
    _pull_seconds = 0.0

    def pull_weights(self):
        """Refresh inference weights from the learner's published copy.
        With a WeightBus on a cuda device this is a device-to-device slice
        copy straight into the prepacked inference tensors (no state_dict,
        no repack, no host round trip); otherwise the reference's
        shared-CPU-model path (worker.py:560-566)."""
        t0 = time.perf_counter()
        if self.weight_bus is not None and self.hip_inf is not None:
            self._bus_ver = self.weight_bus.pull_into(self.hip_inf.pack,
                                                      self._bus_ver)
        else:
            self.model.load_state_dict(self.shared_model.state_dict())
            if self.hip_inf is not None:
                self.hip_inf.refresh()
        self._pulls += 1
        self._pull_seconds += time.perf_counter() - t0
        if self._pulls % 200 == 0:
            print(f"[vector-actor] weight pulls: {self._pulls}, "
                  f"avg {self._pull_seconds / self._pulls * 1e3:.2f} ms "
                  f"({'bus' if self.weight_bus is not None else 'cpu'})",
                  flush=True)
