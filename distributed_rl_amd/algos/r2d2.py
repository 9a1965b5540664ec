"""R2D2 — recurrent replay distributed DQN, MI355X-native.

Behavior parity targets (/root/reference/R2D2/):
  * actor: LSTM policy, per-step hidden state captured BEFORE each step
    (Player.py:30-32,300-301), fixed 80-step sequences with 40-step overlap
    (FIXED_TRAJECTORY=80, emit drops the first 40 on non-terminal emission,
    Player.py:37-62), actor-side sequence priority = eta-mix of n-step
    double-DQN |TD| over the whole window replayed with the stored h_0
    (Player.py:147-215), weight pull every 400 steps (Player.py:321-322),
  * learner: 20-step no-grad burn-in then detachCellState, training on the
    remaining 60 steps (Learner.py:83-107); value rescaling h/h^-1 around
    the n-step bootstrap (Learner.py:22-35,143-167, USE_RESCALING); sequence
    priority (0.9*max + 0.1*mean)^alpha (Learner.py:175-181); Adam(1e-4,
    eps 1e-3), grad clip 40, target sync 2500, publish every 25 steps
    (Learner.py:289-293).

DEFECT FIXED: the reference slices actions with
``action[FIXED_TRAJECTORY - MEM:-1]`` (R2D2/Learner.py:111) which crashes
with the shipped cfg; the correct training window is steps [MEM, T-1) and
that is what we implement: 59 n-step targets per 80-step sequence, truncated
n at the tail (n_t = min(UNROLL_STEP, T-1-t)), bootstrap masked by the
sequence-terminal flag at the final state.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import numpy as np
import torch

from .. import ops
from ..config import Config
from ..models import BaseAgent
from ..replay import make_per, make_r2d2_schema
from .common import LearnerBase

PUBLISH_EVERY = 25  # R2D2/Learner.py:289-293
ACTOR_PULL_EVERY = 400  # R2D2/Player.py:321-322
ETA = 0.9  # priority mix (R2D2 paper / Learner.py:178)


def nstep_recurrent_targets(q_online: torch.Tensor, q_target: torch.Tensor,
                            actions: torch.Tensor, rewards: torch.Tensor,
                            done: torch.Tensor, burn_in: int, n_step: int,
                            gamma: float, use_rescaling: bool):
    """Shared target math for learner loss and actor priorities.

    q_online, q_target: (T, B, A) — full-sequence Q values (q_online may be
    detached or differentiable; only rows [burn_in, T-1) contribute).
    actions/rewards: (T, B); done: (B,) terminal-at-sequence-end flag.
    Returns (td, q_taken, targets) each (T-1-burn_in, B) where
    td = h(G_t) - Q(s_t, a_t) in rescaled space.
    """
    T, B, A = q_online.shape
    h = ops.value_rescale if use_rescaling else (lambda x: x)
    h_inv = ops.inv_value_rescale if use_rescaling else (lambda x: x)

    with torch.no_grad():
        a_star = q_online.detach().argmax(dim=2)  # (T, B)
        q_boot = q_target.gather(2, a_star.unsqueeze(2)).squeeze(2)  # (T, B)
        q_boot = h_inv(q_boot.float())
        # bootstrap at the final state is masked when the sequence terminated
        q_boot = torch.cat(
            [q_boot[:-1], (q_boot[-1] * (1.0 - done)).unsqueeze(0)], dim=0
        )
        ts = torch.arange(burn_in, T - 1, device=q_online.device)
        n_t = torch.minimum(
            torch.full_like(ts, n_step), (T - 1) - ts
        )  # truncated n at the tail
        # discounted n-step return via gamma-weighted prefix sums:
        #   sum_{i<n_t} g^i r_{t+i} = (P[t+n_t] - P[t]) / g^t,
        #   P[t] = sum_{s<t} g^s r_s
        dev = q_online.device
        g_pow = gamma ** torch.arange(T + 1, device=dev, dtype=torch.float64)
        wr = rewards.double() * g_pow[:T].unsqueeze(1)  # (T, B)
        P = torch.zeros(T + 1, B, dtype=torch.float64, device=dev)
        P[1:] = torch.cumsum(wr, dim=0)
        tn = ts + n_t  # (W,)
        ret = (P[tn] - P[ts]) / g_pow[ts].unsqueeze(1)  # (W, B)
        G = ret.float() + (gamma ** n_t.float()).unsqueeze(1) * q_boot[tn]
        targets = h(G)
    q_taken = q_online[burn_in : T - 1].gather(
        2, actions[burn_in : T - 1].long().unsqueeze(2)
    ).squeeze(2).float()
    td = targets - q_taken
    return td, q_taken, targets


class R2D2Learner(LearnerBase):
    ALG = "R2D2"

    def __init__(self, cfg: Config, device: Optional[str] = None, rank: int = 0,
                 world_size: int = 1, transport=None,
                 batch_size: Optional[int] = None,
                 replay_capacity: Optional[int] = None, replay=None,
                 enable_tb: bool = True, run_root: str = "."):
        super().__init__(cfg, device, rank, world_size, run_root=run_root,
                         enable_tb=enable_tb)
        self.batch_size = batch_size or cfg.batch_size
        self.T = cfg.fixed_trajectory
        self.burn_in = cfg.burn_in
        self.n_step = cfg.unroll_step
        self.gamma = cfg.gamma
        self.alpha = cfg.alpha
        self.beta = cfg.beta
        self.use_rescaling = cfg.use_rescaling
        self.model = self.build_model()
        self.mp = None
        # Mixed trunk on GPU: the conv stack runs bf16 channels_last (our
        # MFMA kernels) with fp32 master weights; the LSTM + heads stay fp32
        # end-to-end — measured on MI355X the MIOpen fp32 RNN path beats
        # bf16 nn.LSTM 2.4x, so until the K5 persistent-LSTM HIP kernel
        # lands this split is the fast shape.
        self._bf16_trunk = self.device.type == "cuda"
        if self._bf16_trunk:
            from ..parallel.precision import MixedPrecisionTrainer

            self.model.to(memory_format=torch.channels_last)
            conv_prefixes = tuple(
                f"nodes.{name}." for name, node in self.cfg.model_info.items()
                if str(node.get("netCat", "")).upper() == "CNN2D"
            )
            self.mp = MixedPrecisionTrainer(
                self.model,
                keep_fp32=lambda n, p: not n.startswith(conv_prefixes),
            )
            self.net = self.mp.compute
            import copy as _copy

            self.target = _copy.deepcopy(self.net)
        else:
            self.net = self.model
            self.target = self.build_model()
            self.target.updateParameter(self.model, 1.0)
        for p in self.target.parameters():
            p.requires_grad_(False)
        self.optim = self.build_optim(self.model)
        cap = replay_capacity or cfg.replay_memory_len
        rdev = str(self.device) if self.device.type == "cuda" else "cpu"
        # On GPU, frames are stored NHWC (like Ape-X) so the trunk's
        # channels_last view is free — the NCHW wire layout cost a 92.7 us
        # u8 relayout copy per sequence pass (profiles r7 trace)
        self._nhwc = self.device.type == "cuda" and replay is None
        fshape = (84, 84, 4) if self._nhwc else (4, 84, 84)
        self.replay = replay if replay is not None else make_per(
            cap, make_r2d2_schema(self.T, frame_shape=fshape,
                                  hidden=self._hidden_size()), device=rdev
        )
        self.transport = transport
        self.reducer = None
        self._sqsum_buf = (
            torch.zeros(1, device=self.device) if self.device.type == "cuda" else None
        )

    def _hidden_size(self) -> int:
        for node in self.cfg.model_info.values():
            if str(node.get("netCat", "")).upper() == "LSTMNET":
                return int(node["hiddenSize"])
        raise ValueError("R2D2 cfg has no LSTMNET node")

    # -- ingest -----------------------------------------------------------
    def ingest(self) -> int:
        if self.transport is None:
            return 0
        got = self.transport.drain()
        if got is None:
            return 0
        cols_np, prio_np = got
        cols = {k: torch.from_numpy(np.ascontiguousarray(v))
                for k, v in cols_np.items()}
        prio = torch.from_numpy(np.ascontiguousarray(prio_np))
        if self.device.type == "cuda":
            cols = {k: v.pin_memory().to(self.device, non_blocking=True)
                    for k, v in cols.items()}
            prio = prio.pin_memory().to(self.device, non_blocking=True)
        self.replay.push(self._wire_to_store(cols), prio)
        self.ingested_total += len(prio)
        return len(prio)

    def _wire_to_store(self, cols):
        if getattr(self, "_nhwc", False) and cols["states"].shape[-1] != 4:
            cols = dict(cols)
            cols["states"] = cols["states"].permute(0, 1, 3, 4, 2).contiguous()
        return cols

    def push_sequences(self, cols, prio):
        self.replay.push(self._wire_to_store(cols), prio)

    # -- forward helpers ---------------------------------------------------
    def _seq_forward(self, net: BaseAgent, frames: torch.Tensor, h0, *,
                     burn_in_split: bool):
        """frames: (T, B, 4,84,84) float; returns (T or T-burn_in, B, A) Q.

        burn_in_split=True runs [0, burn_in) under no_grad then detaches the
        cell state (R2D2/Learner.py:99-104) and returns only the training
        window's Q values; False runs the whole sequence (target net)."""
        T, B = frames.shape[:2]
        net.setCellState(h0)

        def prep(chunk, steps):
            if getattr(self, "_nhwc", False) and chunk.shape[-1] == 4:
                # NHWC storage: the channels_last logical-NCHW view is free
                h_, w_ = chunk.shape[2], chunk.shape[3]
                return chunk.reshape(steps * B, h_, w_, 4).permute(0, 3, 1, 2)
            x = chunk.reshape(steps * B, *frames.shape[2:])
            if self._bf16_trunk:
                x = x.contiguous(memory_format=torch.channels_last)
            return x

        if burn_in_split and self.burn_in > 0:
            m = self.burn_in
            with torch.no_grad():
                net.forward([prep(frames[:m], m), torch.tensor([m, B, -1])])
            net.detachCellState()
            q = net.forward(
                [prep(frames[m:], T - m), torch.tensor([T - m, B, -1])]
            )[0]
            return q.view(T - m, B, -1)
        with torch.no_grad():
            q = net.forward([prep(frames, T), torch.tensor([T, B, -1])])[0]
        return q.view(T, B, -1)

    def _h0_to_state(self, h0: torch.Tensor):
        # h0: (B, 2, H) -> ((1,B,H), (1,B,H))
        h = h0[:, 0].unsqueeze(0).contiguous()
        c = h0[:, 1].unsqueeze(0).contiguous()
        return (h.to(self.device), c.to(self.device))

    # -- train -------------------------------------------------------------
    def _fwd_bwd(self, data, weights):
        """Sequence passes + n-step targets + loss + backward + priority
        (collective-free: hipGraph-capturable)."""
        B = data["done"].shape[0]
        T = self.T
        states = data["states"].to(self.device, non_blocking=True)
        if self._bf16_trunk:
            # keep frames uint8, seq-major; the fused conv1 dequants
            # in-kernel. NHWC storage + whole-row transpose kernel: the
            # permute+contiguous relayout cost 92.7 us per pass (r7 trace)
            frames = (ops.seq_transpose(states)
                      if getattr(self, "_nhwc", False)
                      else states.permute(1, 0, 2, 3, 4).contiguous())
        else:
            frames = ops.dequant_frames(
                states.permute(1, 0, 2, 3, 4).contiguous(), torch.float32
            )  # (T,B,4,84,84) seq-major (R2D2/Learner.py:93)
        actions = data["actions"].to(self.device).t().contiguous()  # (T,B)
        rewards = data["rewards"].to(self.device).t().contiguous()
        done = data["done"].to(self.device)
        weights = weights.to(self.device)
        h0 = self._h0_to_state(data["h0"].to(self.device))

        if self.device.type == "cuda":
            # run the TARGET sequence pass concurrently on a side stream:
            # both recurrences are ~32-block persistent/per-step kernels
            # that underfill the 256-CU chip alone, and they are fully
            # independent until the loss. Fork/join records events, so the
            # same structure capture-records as a parallel hipGraph branch.
            if not hasattr(self, "_tgt_stream"):
                self._tgt_stream = torch.cuda.Stream(self.device)
            cur = torch.cuda.current_stream(self.device)
            self._tgt_stream.wait_stream(cur)
            with torch.cuda.stream(self._tgt_stream), torch.no_grad():
                q_tgt_full = self._seq_forward(self.target, frames, h0,
                                               burn_in_split=False)
            q_train = self._seq_forward(self.net, frames, h0,
                                        burn_in_split=True)
            cur.wait_stream(self._tgt_stream)
        else:
            q_train = self._seq_forward(self.net, frames, h0,
                                        burn_in_split=True)
            with torch.no_grad():
                q_tgt_full = self._seq_forward(self.target, frames, h0,
                                               burn_in_split=False)
        if self.device.type == "cuda" and ops.has_r2d2_seq_loss():
            # fused sequence loss: per-(t,b) truncated n-step targets +
            # rescale + double-DQN argmax + IS loss + eta-mix priority in 3
            # kernels (replaces the ~30-launch torch chain incl. an fp64
            # prefix sum)
            loss, prio, value, td_abs = ops.r2d2_sequence_loss(
                q_train.float(), q_tgt_full.float(), actions, rewards, done,
                weights, self.burn_in, self.n_step, self.gamma, self.alpha,
                ETA, self.use_rescaling,
            )
            stats = {"loss": loss.detach(), "value": value, "td_abs": td_abs}
        else:
            # Double-DQN argmax is only consulted at steps t+n >= burn_in+1,
            # all inside the training window, so the burn-in rows of the
            # online view are never read — pad them with (detached) target
            # rows instead of paying a third sequence pass.
            q_online_full = torch.cat([q_tgt_full[: self.burn_in], q_train],
                                      dim=0)
            td, q_taken, targets = nstep_recurrent_targets(
                q_online_full, q_tgt_full, actions, rewards, done,
                self.burn_in, self.n_step, self.gamma, self.use_rescaling,
            )
            # loss: IS-weighted 0.5 * mean_t(td^2) per sequence
            loss = 0.5 * (weights * td.pow(2).mean(dim=0)).mean()
            prio = ops.sequence_priority(td.detach().abs(), self.alpha, ETA)
            stats = {
                "loss": loss.detach(),
                "value": q_taken.detach().mean(),
                "td_abs": td.detach().abs().mean(),
            }

        if self.mp is not None:
            self.mp.direct_grads(loss)
        else:
            self.optim.zero_grad(set_to_none=False)
            loss.backward()
        return stats, prio

    def _optimize_mp(self):
        """Upcast + clip + optimizer + param sync (post-collective stage)."""
        self.mp.upcast_grads()
        self.model.clippingNorm(40.0)  # joint norm over both dtype groups
        self.optim.step()
        self.mp.sync_compute_params()

    def train_step(self, data, idx, weights) -> Dict[str, torch.Tensor]:
        stats, prio = self._fwd_bwd(data, weights)
        if self.mp is not None:
            self.mp.allreduce_grads()
            self._optimize_mp()
        else:
            if self.reducer is not None:
                self.reducer.all_reduce()
            self.model.clippingNorm(40.0)  # R2D2/Learner.py:208
            self.optim.step()
        self.replay.update(idx, prio)
        return stats

    def _inner_step(self):
        data, idx, w = self.replay.sample(self.batch_size, self.beta)
        return self.train_step(data, idx, w)

    def _cadence(self):
        self.step_count += 1
        if self.step_count % self.cfg.target_frequency == 0:
            self.target.load_state_dict(self.model.state_dict())  # casts per tensor
            self.publish_weights(include_target=True)
        elif self.step_count % PUBLISH_EVERY == 0:
            self.publish_weights()

    def step(self):
        stats = self.maybe_profile_first_step(self._inner_step)
        self._cadence()
        return stats

    def make_graphed_step(self, warmup_iters: int = 3):
        """hipGraph-capture the whole R2D2 step (PER sample -> burn-in +
        train + target sequence passes -> n-step rescaled targets -> loss ->
        backward -> Adam -> sequence-priority update). At world_size > 1 the
        step is captured as TWO graphs with the RCCL all-reduce (one per
        dtype group) running eagerly between them — collectives are never
        captured (same structure as ApexLearner.make_graphed_step)."""
        assert self.device.type == "cuda"
        for g in self.optim.param_groups:
            g["capturable"] = True
        split = self.mp is not None and self.mp.world > 1
        side = torch.cuda.Stream(self.device)
        side.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self._inner_step()
        torch.cuda.current_stream(self.device).wait_stream(side)

        if not split:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                static_out = self._inner_step()
            self._graph = graph

            def stepper():
                graph.replay()
                self._cadence()
                return static_out

            return stepper

        # ---- overlapped pipeline (north-star C1): all-reduce on a comm
        # stream runs in parallel with the priority update of batch t + the
        # PER sample of batch t+1 (see ApexLearner.make_graphed_step).
        mp = self.mp
        idx_save = torch.empty(self.batch_size, dtype=torch.int64,
                               device=self.device)
        g_s = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g_s):
            s_data, s_idx, s_w = self.replay.sample(self.batch_size, self.beta)
        g_fb = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g_fb, pool=g_s.pool()):
            static_out, prio = self._fwd_bwd(s_data, s_w)
            idx_save.copy_(s_idx)
        g_upd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g_upd, pool=g_s.pool()):
            self.replay.update(idx_save, prio)
        g_opt = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g_opt, pool=g_s.pool()):
            self._optimize_mp()
        self._graph = (g_s, g_fb, g_upd, g_opt)
        comm = torch.cuda.Stream(self.device)
        g_s.replay()  # prime the first batch

        def stepper():
            cur = torch.cuda.current_stream(self.device)
            g_fb.replay()
            comm.wait_stream(cur)
            with torch.cuda.stream(comm):
                mp.allreduce_grads()
            g_upd.replay()
            g_s.replay()
            cur.wait_stream(comm)
            g_opt.replay()
            self._cadence()
            return static_out

        return stepper

    def make_pipelined_step(self):
        """Eager pipelined step for world_size > 1 without graph capture
        (CPU/gloo rehearsal of the overlap ordering; GPU fallback)."""
        pending = [self.replay.sample(self.batch_size, self.beta)]

        def stepper():
            data, idx, w = pending[0]
            stats, prio = self._fwd_bwd(data, w)
            if self.mp is not None:
                works = self.mp.allreduce_grads_async()
                self.replay.update(idx, prio)
                pending[0] = self.replay.sample(self.batch_size, self.beta)
                for wk in works:
                    wk.wait()
                self._optimize_mp()
            else:
                work = (self.reducer.all_reduce_async()
                        if self.reducer is not None else None)
                self.replay.update(idx, prio)
                pending[0] = self.replay.sample(self.batch_size, self.beta)
                if self.reducer is not None:
                    self.reducer.finish(work)
                self.model.clippingNorm(40.0)
                self.optim.step()
            self._cadence()
            return stats

        return stepper

    # -- weights ------------------------------------------------------------
    def publish_weights(self, include_target: bool = False):
        if self.transport is None or self.rank != 0:
            return
        if self.request_publish(include_target):
            return  # async publisher thread
        self._publish_sync(include_target, self.step_count)

    def _publish_sync(self, include_target: bool, count: int):
        payload = {
            "count": count,
            "state_dict": self.snapshot_state_dict(),
        }
        if include_target:
            payload["target_state_dict"] = {
                k: v.detach().to("cpu", torch.float32)
                for k, v in self.target.state_dict().items()
            }
        self.transport.publish(payload)

    # -- run loop ------------------------------------------------------------
    def run(self, max_steps: int = 1_000_000, warmup_items: Optional[int] = None):
        self.start_ingest_thread()
        need = warmup_items if warmup_items is not None else self.cfg.buffer_size
        t0 = time.time()
        while len(self.replay) <= need:
            if self._ingest_thread is None:
                with self._ingest_lock:
                    self.ingest()
            if time.time() - t0 > 1200:
                raise TimeoutError("R2D2 replay warmup stalled")
            time.sleep(0.01)
        self.publish_weights(include_target=True)
        stepper = None  # hipGraph-captured once the replay ring is full
        while self.step_count < max_steps:
            if self._ingest_thread is None:
                with self._ingest_lock:
                    self.ingest()
            if stepper is None and self.device.type == "cuda" \
                    and len(self.replay) >= self.replay.capacity:
                # n_valid is baked into the captured sample kernel; once the
                # ring is full it stays at capacity, so capture is safe now.
                # Pause the ingest thread: global-mode stream capture forbids
                # concurrent stream work from other threads.
                self.stop_ingest_thread()
                try:
                    stepper = self.make_graphed_step()
                except Exception as e:  # pragma: no cover
                    print(f"[R2D2] graph capture failed ({e}); staying eager",
                          flush=True)
                    stepper = self.step
                self.start_ingest_thread()
            stats = (stepper or self.step)()
            if self.step_count % self.LOG_EVERY == 0:
                rewards = self.transport.drain_rewards() if self.transport else []
                # skip the Reward scalar when nothing was drained (no -21
                # Pong placeholder — see ApexLearner._log_block)
                mean_r = float(np.mean(rewards)) if rewards else float("nan")
                if rewards:
                    self.log_scalar("Reward", mean_r)
                self.log_scalar("value", float(stats["value"]))
                self.log_scalar("norm", float(self.model.calculateNorm()))
                if self.rank == 0:
                    print(
                        f"[R2D2] step={self.step_count} loss={float(stats['loss']):.5f} "
                        f"value={float(stats['value']):.3f} reward={mean_r:.1f} "
                        f"replay={len(self.replay)}", flush=True,
                    )
            if self.step_count % self.CKPT_EVERY == 0:
                self.save_checkpoint()

    # -- checkpoint ----------------------------------------------------------
    def state_for_checkpoint(self):
        return {
            "alg": self.ALG,
            "model": self.model.state_dict(),
            "target": self.target.state_dict(),
            "optim": self.optim.state_dict(),
            "step": self.step_count,
        }

    def load_from_checkpoint(self, state):
        self.model.load_state_dict(state["model"])
        self.target.load_state_dict(state["target"])
        self.optim.load_state_dict(state["optim"])
        self.step_count = int(state["step"])
        if self.mp is not None:
            self.mp.sync_compute_params()

    def load_model_only(self, sd):
        self.model.load_state_dict(sd)
        self.target.load_state_dict(sd)
        if self.mp is not None:
            self.mp.sync_compute_params()


# ===========================================================================
# Actor
# ===========================================================================


class R2D2Player:
    """Recurrent actor: overlapping 80-step sequences with stored initial
    hidden state and actor-side sequence priorities."""

    def __init__(self, cfg: Config, idx: int, transport, env=None,
                 env_kind: str = "auto", seed: Optional[int] = None):
        from ..actors.env import make_env

        self.cfg = cfg
        self.idx = idx
        self.transport = transport
        self.env = env or make_env(
            env_kind, seed=seed if seed is not None else idx,
            reward_clip=cfg.use_reward_clip,
        )
        self.model = BaseAgent(cfg.model_info).to(cfg.actor_device).eval()
        self.target = BaseAgent(cfg.model_info).to(cfg.actor_device).eval()
        n_actors = max(cfg.num_actors, 2)
        self.eps = 0.4 ** (1 + 7 * idx / (n_actors - 1))
        self.T = cfg.fixed_trajectory
        self.overlap = self.T // 2  # 40-step overlap (Player.py:60-61)
        self.n_step = cfg.unroll_step
        self.gamma = cfg.gamma
        self.alpha = cfg.alpha
        self.action_n = cfg.action_size
        self.use_rescaling = cfg.use_rescaling
        self.rng = np.random.default_rng(3000 + idx)
        self.env_steps = 0
        self.weight_version = -1
        self._steps: List[tuple] = []  # (state, action, reward)
        self._hiddens: List[tuple] = []  # cell state BEFORE each step
        # rolling last-T history for the terminal emission
        # (R2D2/Player.py:37-47 emits storage[-3*FT:] with done=True)
        from collections import deque as _deque

        self._recent: _deque = _deque(maxlen=self.T)
        self._recent_hid: _deque = _deque(maxlen=self.T)

    # -- inference ---------------------------------------------------------
    @torch.no_grad()
    def act(self, state_u8: np.ndarray) -> int:
        x = torch.from_numpy(state_u8).unsqueeze(0).float() / 255.0
        q = self.model.forward([x, torch.tensor([1, 1, -1])])[0]
        if self.rng.random() < self.eps:
            return int(self.rng.integers(0, self.action_n))
        return int(q.argmax(1).item())

    # -- sequence priority (replay window under no_grad) -------------------
    @torch.no_grad()
    def _sequence_priority(self, states_np, actions_np, rewards_np,
                           h0_state, done: float) -> float:
        T = self.T
        frames = torch.from_numpy(states_np).float().unsqueeze(1) / 255.0  # (T,1,...)
        live_m = self.model.getCellState()
        live_t = self.target.getCellState()
        self.model.setCellState((h0_state[0].clone(), h0_state[1].clone()))
        self.target.setCellState((h0_state[0].clone(), h0_state[1].clone()))
        hint = torch.tensor([T, 1, -1])
        q_on = self.model.forward([frames.reshape(T, *frames.shape[2:]), hint]
                                  )[0].view(T, 1, -1)
        q_tg = self.target.forward([frames.reshape(T, *frames.shape[2:]), hint]
                                   )[0].view(T, 1, -1)
        self.model.setCellState(live_m)
        self.target.setCellState(live_t)
        td, _, _ = nstep_recurrent_targets(
            q_on, q_tg, torch.from_numpy(actions_np).unsqueeze(1),
            torch.from_numpy(rewards_np).unsqueeze(1),
            torch.tensor([done]), 0, self.n_step, self.gamma,
            self.use_rescaling,
        )
        mix = ETA * td.abs().max() + (1 - ETA) * td.abs().mean()
        return float(mix ** self.alpha)

    def _emit(self, steps, h0_state, done: bool):
        states_np = np.stack([s[0] for s in steps])
        actions_np = np.array([s[1] for s in steps], np.int32)
        rewards_np = np.array([s[2] for s in steps], np.float32)
        prio = self._sequence_priority(states_np, actions_np, rewards_np,
                                       h0_state, 1.0 if done else 0.0)
        h0_np = torch.cat([h0_state[0][0], h0_state[1][0]], dim=0).numpy()  # (2,H)
        cols = {
            "h0": h0_np[None],
            "states": states_np[None],
            "actions": actions_np[None],
            "rewards": rewards_np[None],
            "done": np.array([1.0 if done else 0.0], np.float32),
        }
        self.transport.push(cols, np.array([prio], np.float32))

    def pull_weights(self):
        payload = self.transport.fetch()
        if payload is None:
            return
        if payload.get("count", 0) == self.weight_version:
            return
        self.model.load_state_dict(payload["state_dict"])
        if "target_state_dict" in payload:
            self.target.load_state_dict(payload["target_state_dict"])
        self.weight_version = payload.get("count", 0)

    def _zero_hidden(self):
        self.model.zeroCellState(1)
        self.target.zeroCellState(1)

    def run(self, max_env_steps: int = 1_000_000):
        self.pull_weights()
        self._zero_hidden()
        state = self.env.reset()
        episode_reward = 0.0
        while self.env_steps < max_env_steps:
            h = self.model.getCellState()
            hid = (h[0].clone(), h[1].clone())
            self._hiddens.append(hid)
            self._recent_hid.append(hid)
            action = self.act(state)
            next_state, reward, done, info = self.env.step(action)
            episode_reward += reward
            self._steps.append((state, action, reward))
            self._recent.append((state, action, reward))
            if len(self._steps) == self.T:
                self._emit(self._steps[: self.T], self._hiddens[0], done)
                if done:
                    self._steps, self._hiddens = [], []
                else:
                    self._steps = self._steps[self.overlap:]
                    self._hiddens = self._hiddens[self.overlap:]
            elif done and len(self._recent) == self.T:
                # terminal emission: the LAST T steps ending at the terminal
                # step (R2D2/Player.py:37-47) — without it, episodes whose
                # length is not a window boundary never produce a done=1
                # sequence and the value function never sees termination
                self._emit(list(self._recent), self._recent_hid[0], True)
                self._steps, self._hiddens = [], []
            elif done:
                self._steps, self._hiddens = [], []
            state = next_state
            self.env_steps += 1
            if self.env_steps % ACTOR_PULL_EVERY == 0:
                self.pull_weights()
            if done:
                self.transport.push_reward(self.idx, episode_reward, self.eps)
                episode_reward = 0.0
                self._zero_hidden()
                self._recent.clear()
                self._recent_hid.clear()
                state = self.env.reset()
