"""IMPALA — V-trace actor-critic, MI355X-native.

Behavior parity targets (/root/reference/IMPALA/):
  * actor: single net, 7 outputs = 6 logits + 1 value (cfg/impala.json:44),
    Categorical sampling, behavior probability mu(a|s) recorded per step
    (Player.py:146-148,202-204), 20-step unrolls + bootstrap state +
    not_done flag (Player.py:181-196), short terminal unrolls back-filled
    from the previous unroll (checkLength, Player.py:116-125),
  * learner: V-trace with clipped rho/c (rho_bar=P_VALUE, c_bar=C_VALUE,
    lam=C_LAMBDA; Learner.py:151-213), losses
    actor = mean(log pi(a) * adv) + ENTROPY_R * entropy, critic = 0.5*MSE,
    total = -actor + critic (Learner.py:95-119,224), grad clip 40
    (Learner.py:259), weights published every train step (Learner.py:286-287),
    actors pull every 400 env steps (Player.py:198-199).

Design delta: the reference runs TWO forward passes per batch (no-grad
target construction + differentiable calLoss re-forward, Learner.py:121-222).
We run ONE differentiable forward and detach its outputs for the V-trace
targets — mathematically identical (same parameters) at half the FLOPs.
The V-trace scan itself is the K8 HIP kernel on GPU.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import numpy as np
import torch
import torch.nn.functional as F

from .. import ops
from ..config import Config
from ..models import BaseAgent
from ..replay import FifoReplay
from .common import LearnerBase

ACTOR_PULL_EVERY = 400  # IMPALA/Player.py:198-199


def make_impala_schema(unroll: int = 20, frame_shape=(4, 84, 84)):
    return {
        "states": ((unroll + 1, *frame_shape), torch.uint8),
        "actions": ((unroll,), torch.int32),
        "mu": ((unroll,), torch.float32),  # behavior policy prob of action
        "rewards": ((unroll,), torch.float32),
        "not_done": ((), torch.float32),
    }


class ImpalaLearner(LearnerBase):
    ALG = "IMPALA"
    CKPT_EVERY = 100  # IMPALA/Learner.py:290-297

    def __init__(self, cfg: Config, device: Optional[str] = None, rank: int = 0,
                 world_size: int = 1, transport=None,
                 batch_size: Optional[int] = None,
                 replay_capacity: Optional[int] = None, replay=None,
                 publish_every: int = 1, enable_tb: bool = True,
                 run_root: str = "."):
        super().__init__(cfg, device, rank, world_size, run_root=run_root,
                         enable_tb=enable_tb)
        self.batch_size = batch_size or cfg.batch_size
        self.unroll = cfg.unroll_step
        self.model = self.build_model()
        self.mp = None
        if self.device.type == "cuda":
            from ..parallel.precision import MixedPrecisionTrainer

            self.model.to(memory_format=torch.channels_last)
            self.mp = MixedPrecisionTrainer(self.model)
            self.net = self.mp.compute
        else:
            self.net = self.model
        self.optim = self.build_optim(self.model)
        cap = replay_capacity or cfg.replay_memory_len
        rdev = str(self.device) if self.device.type == "cuda" else "cpu"
        self.replay = replay if replay is not None else FifoReplay(
            cap, make_impala_schema(self.unroll), device=rdev)
        self.transport = transport
        self.publish_every = publish_every
        self.gamma = cfg.gamma
        # Replay-reuse cap (cfg MAX_REPLAY_REUSE, default off = reference
        # behavior): bound consumed/ingested trajectories so a fast learner
        # cannot spin on stale FIFO contents — the round-1 live run showed
        # entropy collapse after ~2k steps when the learner outran the fleet
        # (profiles/r01_learning_sanity.md); the reference is implicitly
        # paced by its slow Redis pipe, this makes the ratio explicit.
        self.max_replay_reuse = float(cfg.get("MAX_REPLAY_REUSE", 0) or 0)
        self.reducer = None
        self._sqsum_buf = (
            torch.zeros(1, device=self.device) if self.device.type == "cuda" else None
        )

    # -- ingest ----------------------------------------------------------
    def ingest(self) -> int:
        if self.transport is None:
            return 0
        got = self.transport.drain()
        if got is None:
            return 0
        cols_np, _ = got
        cols = {
            k: torch.from_numpy(np.ascontiguousarray(v)) for k, v in cols_np.items()
        }
        n = cols["not_done"].shape[0]
        if self.device.type == "cuda":
            cols = {k: v.pin_memory().to(self.device, non_blocking=True)
                    for k, v in cols.items()}
        self.replay.push(cols)
        self.ingested_total += n
        return n

    def push_trajectories(self, cols: Dict[str, torch.Tensor]):
        self.replay.push(cols)

    # -- train -----------------------------------------------------------
    def _fwd_bwd(self, data) -> Dict[str, torch.Tensor]:
        """Forward + V-trace + fused loss + backward (grads left in the
        flat buffers). Collective-free, so it is hipGraph-capturable."""
        cuda = self.device.type == "cuda"
        T = self.unroll
        B = data["not_done"].shape[0]
        states = data["states"].to(self.device, non_blocking=True)
        flat = states.reshape(B * (T + 1), *states.shape[2:])
        if cuda:
            x = ops.dequant_frames_nhwc(flat)
        else:
            x = ops.dequant_frames(flat, torch.float32)
        out = self.net.forward([x])[0].float()  # (B*(T+1), A+1)
        A = out.shape[1] - 1
        logits = out[:, :A].view(B, T + 1, A)
        values = out[:, A].view(B, T + 1)

        actions = data["actions"].to(self.device).long()  # (B, T)
        mu = data["mu"].to(self.device).clamp_min(1e-8)  # behavior probs
        rewards = data["rewards"].to(self.device)
        not_done = data["not_done"].to(self.device)

        v_t = values[:, :T]
        bootstrap = values[:, T]
        if cuda:
            # K9 fused path: one softmax-stats kernel feeds V-trace; the
            # whole loss keeps only `out` in the autograd graph (the fused
            # backward writes d out in a single launch), so every slice /
            # reshape below is detached
            logits_flat = (logits[:, :T].detach().reshape(B * T, A)
                           .contiguous())
            actions_flat = actions.reshape(-1)
            stats9 = ops.policy_softmax_stats(logits_flat, actions_flat)
            log_pi_a_det = stats9[0].view(B, T)
            v_t_det = v_t.detach().contiguous()
        else:
            log_pi = F.log_softmax(logits[:, :T], dim=-1)  # (B, T, A)
            log_pi_a = log_pi.gather(2, actions.unsqueeze(2)).squeeze(2)
            log_pi_a_det = log_pi_a.detach()
            v_t_det = v_t.detach()

        with torch.no_grad():
            # (B,T)-native V-trace: no transpose/log round-trips on GPU
            vs, pg_adv = ops.vtrace_bt(
                mu, log_pi_a_det, rewards, v_t_det, bootstrap.detach(),
                not_done, self.gamma, rho_bar=self.cfg.p_value,
                c_bar=self.cfg.c_value, lam=self.cfg.c_lambda,
            )

        if cuda:
            # one-kernel total loss (pg objective + entropy bonus + critic
            # MSE) + one-kernel whole-head backward — replaces the
            # ~10-launch torch reduction chain and both SliceBackwards
            loss, obj_actor, critic_loss = ops.impala_fused_loss(
                out, v_t_det, stats9, actions_flat, pg_adv.reshape(-1),
                vs, self.cfg.entropy_r, T,
            )
            entropy = stats9[3]
        else:
            pi = log_pi.exp()
            entropy = -(pi * log_pi).sum(-1).mean()
            obj_actor = (log_pi_a * pg_adv).mean() + self.cfg.entropy_r * entropy
            critic_loss = 0.5 * F.mse_loss(v_t, vs)
            loss = -obj_actor + critic_loss

        if self.mp is not None:
            self.mp.direct_grads(loss)
        else:
            self.optim.zero_grad(set_to_none=False)
            loss.backward()
        return {
            "loss": loss.detach(),
            "obj_actor": obj_actor.detach(),
            "critic_loss": critic_loss.detach(),
            "entropy": entropy.detach(),
            # telemetry means are computed lazily at log time (outside any
            # captured graph) from these static buffers — 3 fewer reduce
            # kernels in the per-step hot path
            "v_t": v_t.detach(),
            "vs": vs,
            "pg_adv": pg_adv,
        }

    def _optimize_mp(self):
        """Upcast + clip + optimizer + param sync (post-collective stage;
        hipGraph-capturable)."""
        self.mp.upcast_grads()
        ops.clip_flat_grad_(self.mp.flat_mgrad, 40.0, self._sqsum_buf)
        self.optim.step()
        self.mp.sync_compute_params()

    def train_step(self, data) -> Dict[str, torch.Tensor]:
        stats = self._fwd_bwd(data)
        if self.mp is not None:
            self.mp.allreduce_grads()
            self._optimize_mp()
        else:
            if self.reducer is not None:
                self.reducer.all_reduce()
            self.model.clippingNorm(40.0)
            self.optim.step()
        return stats

    def _inner_step(self):
        data, _, _ = self.replay.sample(self.batch_size)
        return self.train_step(data)

    def _cadence(self):
        self.step_count += 1
        if self.step_count % self.publish_every == 0:
            self.publish_weights()

    def step(self):
        stats = self.maybe_profile_first_step(self._inner_step)
        self._cadence()
        return stats

    def make_graphed_step(self, warmup_iters: int = 3):
        """hipGraph-capture the learner step. At world_size > 1 the step is
        captured as TWO graphs with the RCCL all-reduce running eagerly
        between them — collectives are never captured (same structure as
        ApexLearner.make_graphed_step)."""
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

        # ---- overlapped pipeline (north-star C1): the RCCL all-reduce on a
        # comm stream runs in parallel with the NEXT batch's FIFO sample on
        # the compute stream (see ApexLearner.make_graphed_step).
        mp = self.mp
        g_s = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g_s):
            s_data, _, _ = self.replay.sample(self.batch_size)
        g_fb = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g_fb, pool=g_s.pool()):
            static_out = self._fwd_bwd(s_data)
        g_opt = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g_opt, pool=g_s.pool()):
            self._optimize_mp()
        self._graph = (g_s, g_fb, g_opt)
        comm = torch.cuda.Stream(self.device)
        g_s.replay()  # prime the first batch

        def stepper():
            cur = torch.cuda.current_stream(self.device)
            g_fb.replay()
            comm.wait_stream(cur)
            with torch.cuda.stream(comm):
                mp.allreduce_grads()
            g_s.replay()  # next sample overlaps the collective
            cur.wait_stream(comm)
            g_opt.replay()
            self._cadence()
            return static_out

        return stepper

    def make_pipelined_step(self):
        """Eager pipelined step for world_size > 1 without graph capture
        (CPU/gloo rehearsal of the overlap ordering; GPU fallback)."""
        pending = [self.replay.sample(self.batch_size)]

        def stepper():
            data, _, _ = pending[0]
            stats = self._fwd_bwd(data)
            if self.mp is not None:
                works = self.mp.allreduce_grads_async()
                pending[0] = self.replay.sample(self.batch_size)
                for wk in works:
                    wk.wait()
                self._optimize_mp()
            else:
                work = (self.reducer.all_reduce_async()
                        if self.reducer is not None else None)
                pending[0] = self.replay.sample(self.batch_size)
                if self.reducer is not None:
                    self.reducer.finish(work)
                self.model.clippingNorm(40.0)
                self.optim.step()
            self._cadence()
            return stats

        return stepper

    # -- weights ----------------------------------------------------------
    def publish_weights(self, include_target: bool = False):
        if self.transport is None or self.rank != 0:
            return
        if self.request_publish(include_target):
            return  # async publisher thread (IMPALA publishes EVERY step)
        self._publish_sync(include_target, self.step_count)

    def _publish_sync(self, include_target: bool, count: int):
        self.transport.publish({"count": count,
                                "state_dict": self.snapshot_state_dict()})

    # -- run loop (per-step TB scalars; SURVEY §5.5: 9 scalars per step) ---
    def run(self, max_steps: int = 1_000_000, warmup_items: Optional[int] = None):
        self.start_ingest_thread()
        need = warmup_items if warmup_items is not None else self.batch_size
        t0 = time.time()
        while len(self.replay) < need:
            if self._ingest_thread is None:
                with self._ingest_lock:
                    self.ingest()
            if time.time() - t0 > 600:
                raise TimeoutError("IMPALA replay warmup stalled")
            time.sleep(0.01)
        self.publish_weights()
        stepper = None  # hipGraph-captured once the FIFO ring is full
        while self.step_count < max_steps:
            if self._ingest_thread is None:
                with self._ingest_lock:
                    self.ingest()
            if self.max_replay_reuse > 0 and self.transport is not None:
                # block until the fleet has produced enough fresh unrolls
                t_gate = time.time()
                while ((self.step_count + 1) * self.batch_size
                       > self.max_replay_reuse * max(self.ingested_total, 1)):
                    if self._ingest_thread is None:
                        with self._ingest_lock:
                            self.ingest()
                    time.sleep(0.002)
                    if time.time() - t_gate > 600:
                        raise TimeoutError("IMPALA reuse gate starved "
                                           "(actors dead?)")
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
                    print(f"[IMPALA] graph capture failed ({e}); staying eager",
                          flush=True)
                    stepper = self.step
                self.start_ingest_thread()
            t1 = time.perf_counter()
            stats = (stepper or self.step)()
            dt = time.perf_counter() - t1
            rewards = self.transport.drain_rewards() if self.transport else []
            mean_r = float(np.mean(rewards)) if rewards else None
            if self.writer is not None:
                if mean_r is not None:
                    self.log_scalar("Reward", mean_r)
                self.log_scalar("Objective of Actor", float(stats["obj_actor"]))
                self.log_scalar("Loss of Critic", float(stats["critic_loss"]))
                self.log_scalar("Entropy", float(stats["entropy"]))
                value = float(stats["v_t"].mean())
                target_value = float(stats["vs"].mean())
                self.log_scalar("Advantage", float(stats["pg_adv"].mean()))
                self.log_scalar("Target Value", target_value)
                self.log_scalar("Value", value)
                self.log_scalar("Target_minus_value", target_value - value)
                self.log_scalar("training_Time", dt)
                self.log_scalar("Norm of Gradient", float(self.model.calculateNorm()))
            if mean_r is not None:
                self._last_reward = mean_r
            if self.rank == 0 and self.step_count % 500 == 0:
                print(
                    f"[IMPALA] step={self.step_count} "
                    f"loss={float(stats['loss']):.5f} "
                    f"entropy={float(stats['entropy']):.3f} "
                    f"value={float(stats['v_t'].mean()):.3f} "
                    f"reward={getattr(self, '_last_reward', float('nan')):.1f} "
                    f"replay={len(self.replay)}",
                    flush=True,
                )
            if self.step_count % self.CKPT_EVERY == 0:
                self.save_checkpoint()
        return None

    # -- checkpoint --------------------------------------------------------
    def state_for_checkpoint(self):
        return {
            "alg": self.ALG,
            "model": self.model.state_dict(),
            "optim": self.optim.state_dict(),
            "step": self.step_count,
        }

    def load_from_checkpoint(self, state):
        self.model.load_state_dict(state["model"])
        self.optim.load_state_dict(state["optim"])
        self.step_count = int(state["step"])
        if self.mp is not None:
            self.mp.sync_compute_params()

    def load_model_only(self, sd):
        self.model.load_state_dict(sd)
        if self.mp is not None:
            self.mp.sync_compute_params()


# ===========================================================================
# Actor
# ===========================================================================


class ImpalaPlayer:
    """On-policy actor: 20-step unroll assembly with bootstrap state and
    past-buffer back-fill for short terminal unrolls."""

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
        self.unroll = cfg.unroll_step
        self.action_n = cfg.action_size
        self.env_steps = 0
        self.weight_version = -1
        self.rng = np.random.default_rng(2000 + idx)
        self._steps: List[tuple] = []  # (state, action, mu, reward)
        self._prev_steps: List[tuple] = []  # last full unroll (for back-fill)

    @torch.no_grad()
    def act(self, state_u8: np.ndarray):
        x = torch.from_numpy(state_u8).unsqueeze(0).float() / 255.0
        out = self.model.forward([x])[0][0]
        logits = out[: self.action_n]
        probs = torch.softmax(logits, dim=-1).numpy().astype(np.float64)
        probs = probs / probs.sum()
        a = int(self.rng.choice(self.action_n, p=probs))
        return a, float(probs[a])

    def _emit(self, steps: List[tuple], bootstrap_state, not_done: float):
        T = self.unroll
        states = np.stack([s[0] for s in steps] + [bootstrap_state])
        cols = {
            "states": states[None],  # (1, T+1, 4,84,84)
            "actions": np.array([[s[1] for s in steps]], np.int32),
            "mu": np.array([[s[2] for s in steps]], np.float32),
            "rewards": np.array([[s[3] for s in steps]], np.float32),
            "not_done": np.array([not_done], np.float32),
        }
        self.transport.push(cols, None)

    def pull_weights(self):
        payload = self.transport.fetch()
        if payload is None:
            return
        if payload.get("count", 0) == self.weight_version:
            return
        self.model.load_state_dict(payload["state_dict"])
        self.weight_version = payload.get("count", 0)

    def run(self, max_env_steps: int = 1_000_000):
        self.pull_weights()
        state = self.env.reset()
        episode_reward = 0.0
        while self.env_steps < max_env_steps:
            action, mu = self.act(state)
            next_state, reward, done, _ = self.env.step(action)
            episode_reward += reward
            self._steps.append((state, action, mu, reward))
            if len(self._steps) == self.unroll:
                self._emit(self._steps, next_state, 0.0 if done else 1.0)
                self._prev_steps = self._steps
                self._steps = []
            elif done:
                # back-fill short terminal unroll from the previous unroll
                # (IMPALA/Player.py:116-125 checkLength semantics)
                short = len(self._steps)
                if self._prev_steps:
                    pad = self._prev_steps[-(self.unroll - short):]
                    self._emit(list(pad) + self._steps, next_state, 0.0)
                self._steps = []
            state = next_state
            self.env_steps += 1
            if self.env_steps % ACTOR_PULL_EVERY == 0:
                self.pull_weights()
            if done:
                self.transport.push_reward(self.idx, episode_reward)
                episode_reward = 0.0
                state = self.env.reset()
