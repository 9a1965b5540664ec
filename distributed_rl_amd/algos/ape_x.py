"""Ape-X DQN — MI355X-native actor-learner implementation.

Behavior parity targets (/root/reference/APE_X/):
  * actor: per-actor eps ladder 0.4^(1+7*i/(N-1)) (Player.py:78), n-step
    transition assembly with UNROLL_STEP and gamma folding (Player.py:38-51),
    actor-side initial priority = clipped double-DQN TD error ^ALPHA
    (Player.py:135-159), weight pull every 100 env steps (Player.py:263-264),
  * learner: n-step double-DQN target with (1-done) mask (Learner.py:83-103),
    TD clip [-1,1], new priority (|td|+1e-7)^alpha, loss 0.5*mean(w*td^2)
    (Learner.py:106-114), target hard-sync every TARGET_FREQUENCY steps,
    online weight publish every 50 steps (Learner.py:207-216).
  NOTE: the reference hardcodes gamma=0.99 in the bootstrap (Learner.py:103)
  — a defect; we use cfg GAMMA (identical for the shipped cfg).

Architecture deltas (the MI355X redesign): replay is device-resident
(HipSumTreePER in HBM3E), ingest is pinned-staging hipMemcpyAsync on a side
stream, the loss/priority math is one fused kernel (K4), and the learner can
run data-parallel across GPUs with RCCL all-reduce (parallel/ddp.py).
"""

from __future__ import annotations

import time
from collections import deque
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from .. import ops
from ..config import Config
from ..models import BaseAgent
from ..replay import make_apex_schema, make_per
from .common import LearnerBase

PUBLISH_EVERY = 50  # APE_X/Learner.py:212-216
ACTOR_PULL_EVERY = 100  # APE_X/Player.py:263-264


class ApexLearner(LearnerBase):
    ALG = "APE_X"

    def __init__(self, cfg: Config, device: Optional[str] = None, rank: int = 0,
                 world_size: int = 1, transport=None,
                 batch_size: Optional[int] = None,
                 replay_capacity: Optional[int] = None,
                 replay_device: Optional[str] = None,
                 replay_state_dtype: Optional[torch.dtype] = None,
                 replay=None, enable_tb: bool = True, run_root: str = "."):
        super().__init__(cfg, device, rank, world_size, run_root=run_root,
                         enable_tb=enable_tb)
        self.batch_size = batch_size or cfg.batch_size
        self.model = self.build_model()
        self.mp = None
        if self.device.type == "cuda":
            # channels_last convs (NHWC is MIOpen/our-conv native layout) +
            # persistent bf16 compute replica with flat grad/param buffers
            from ..parallel.precision import MixedPrecisionTrainer

            self.model.to(memory_format=torch.channels_last)
            # pin the two dueling streams' first-layer params adjacent in
            # the flat buffer so both streams run as ONE fused GEMM
            order = self._dueling_param_order()
            self.mp = MixedPrecisionTrainer(self.model, param_order=order)
            self.net = self.mp.compute  # bf16 forward/backward model
            import copy as _copy

            self.target = _copy.deepcopy(self.mp.compute)
            for p in self.target.parameters():
                p.requires_grad_(False)
            # flat target param buffer so hard-sync is ONE device copy
            # (SAME param ordering as the compute flat buffer)
            from ..parallel.precision import _to_flat, _view_like

            tnamed = dict(self.target.named_parameters())
            tp = [tnamed[n] for n in self.mp._name_order]
            self.flat_tparam = torch.empty(
                sum(p.numel() for p in tp), dtype=torch.bfloat16,
                device=self.device,
            )
            off = 0
            with torch.no_grad():
                for p in tp:
                    n = p.numel()
                    self.flat_tparam[off : off + n].copy_(_to_flat(p))
                    p.data = _view_like(self.flat_tparam[off : off + n], p)
                    off += n
            self._build_fast_forward(order)
        else:
            self.net = self.model
            self.target = self.build_model()
            self.target.updateParameter(self.model, 1.0)
            for p in self.target.parameters():
                p.requires_grad_(False)
        self.optim = self.build_optim(self.model)
        cap = replay_capacity or cfg.replay_memory_len
        rdev = replay_device or (
            str(self.device) if self.device.type == "cuda" else "cpu"
        )
        # On GPU, frames are stored NHWC so the first fused conv consumes the
        # uint8 replay column directly (fused dequant in conv_mfma.hip) —
        # no standalone dequant pass, no layout transposes.
        self._nhwc = self.device.type == "cuda"
        # Replay frame storage dtype: uint8 (default; 4x denser than fp16)
        # or float16 (BASELINE config 5's "fp16 replay compression" option —
        # frames stored pre-scaled to [0,1]).
        self.state_dtype = replay_state_dtype or torch.uint8
        if replay is not None:
            # injected replay (e.g. replay.server.RemoteReplay, 3-tier mode);
            # batches arrive NCHW on the replay's own device
            self.replay = replay
            self._nhwc = False
        else:
            schema = (
                make_apex_schema(frame_shape=(84, 84, 4),
                                 state_dtype=self.state_dtype)
                if self._nhwc else make_apex_schema(state_dtype=self.state_dtype)
            )
            self.replay = make_per(cap, schema, device=rdev)
        self.transport = transport
        self.gamma = cfg.gamma
        self.n_step = cfg.unroll_step
        self.alpha = cfg.alpha
        self.beta = cfg.beta
        self.reducer = None  # set by parallel.ddp.attach() for world_size > 1
        self._ingest_stream = (
            torch.cuda.Stream(self.device) if self.device.type == "cuda" else None
        )
        self._staging = None  # lazily allocated pinned buffers

    # ------------------------------------------------------------------
    # fused dueling-stream forward: both 3136->512 stream GEMMs run as ONE
    # (B,3136)x(3136,1024) GEMM through adjacent flat-buffer views
    # ------------------------------------------------------------------
    def _dueling_param_order(self) -> List[str]:
        d = getattr(self.model, "_dueling", {})
        if len(d) != 1:
            return []
        (a_node, v_node), = d.values()
        if a_node not in self.model.nodes or v_node not in self.model.nodes:
            return []
        a_mod = self.model.nodes[a_node]
        v_mod = self.model.nodes[v_node]
        try:
            ok = (a_mod.body[0].in_features == v_mod.body[0].in_features
                  and a_mod.body[0].out_features == v_mod.body[0].out_features
                  and len(a_mod.body) == 4)
        except (AttributeError, IndexError, TypeError):
            ok = False
        if not ok:
            return []
        self._fast_nodes = (a_node, v_node)
        return [
            f"nodes.{a_node}.body.0.weight", f"nodes.{v_node}.body.0.weight",
            f"nodes.{a_node}.body.0.bias", f"nodes.{v_node}.body.0.bias",
        ]

    def _build_fast_forward(self, order: List[str]):
        self._fast_fwd = None
        self._fast_heads = None
        if not order:
            return
        import torch.nn.functional as F

        a_node, v_node = self._fast_nodes
        a_mod = self.net.nodes[a_node]
        hidden = a_mod.body[0].out_features
        in_f = a_mod.body[0].in_features
        # leaf views over the flat compute buffers so autograd deposits
        # grads straight into flat_cgrad
        w1 = self.mp.flat_view(order[:2]).view(2 * hidden, in_f
                                               ).detach().requires_grad_(True)
        w1.grad = self.mp.flat_view(order[:2], "grad").view_as(w1)
        b1 = self.mp.flat_view(order[2:]).detach().requires_grad_(True)
        b1.grad = self.mp.flat_view(order[2:], "grad").view_as(b1)
        a_head = self.net.nodes[a_node].body[2]
        v_head = self.net.nodes[v_node].body[2]
        cnn_node = next(
            n for n, c in self.cfg.model_info.items()
            if str(c.get("netCat", "")).upper() == "CNN2D"
        )
        cnn = self.net.nodes[cnn_node]

        import os as _os

        # own linear_relu kernel measured SLOWER than hipBLASLt at this
        # shape even after prefetch pipelining (A/B: 0.741 vs 0.576 ms
        # step) — the 49-chunk stage/sync cadence dominates; dispatch keeps
        # the library GEMM, DRL_OWN_LINEAR=1 forces ours
        own_lin = (ops.linear_relu_supported(in_f, 2 * hidden)
                   and _os.environ.get("DRL_OWN_LINEAR", "0") == "1")

        def hidden_of(feat):
            if own_lin and feat.dtype == torch.bfloat16:
                return ops.fused_linear_relu(feat, w1, b1)
            return F.relu(F.linear(feat.to(w1.dtype), w1, b1))

        def fast_fwd(x):
            h = hidden_of(cnn(x))
            adv = a_head(h[:, :hidden])
            val = v_head(h[:, hidden:])
            return ops.dueling_head(adv.float(), val.float())

        self._fast_fwd = fast_fwd

        def fast_heads(x):
            h = hidden_of(cnn(x))
            return a_head(h[:, :hidden]), v_head(h[:, hidden:])

        self._fast_heads = fast_heads

        def fast_hidden(x):
            return hidden_of(cnn(x))

        self._fast_hidden = fast_hidden
        # direct-grad leaf list: the forward uses the FUSED (w1, b1) views
        # instead of the four pinned params, and w1/b1's flattened grads
        # cover exactly the flat buffer's pinned prefix
        self._direct_list = [w1, b1] + list(self.mp.groups[0].c_params[4:])
        self._head_params = (a_head.weight, a_head.bias, v_head.weight,
                             v_head.bias)
        self._use_q_loss = ops.has_dueling_q_loss(hidden, self.cfg.action_size)
        # target-net twin over the flat target buffer (same pinned-front
        # ordering as the compute flat buffer, so the same offsets hold)
        t_cnn = self.target.nodes[cnn_node]
        ta_head = self.target.nodes[a_node].body[2]
        tv_head = self.target.nodes[v_node].body[2]
        w1t = self.flat_tparam[: 2 * hidden * in_f].view(2 * hidden, in_f)
        b1t = self.flat_tparam[2 * hidden * in_f : 2 * hidden * (in_f + 1)]

        def t_hidden_of(feat):
            if own_lin and feat.dtype == torch.bfloat16:
                return ops.fused_linear_relu(feat, w1t, b1t)
            return F.relu(F.linear(feat.to(w1t.dtype), w1t, b1t))

        def target_heads(x):
            h = t_hidden_of(t_cnn(x))
            return ta_head(h[:, :hidden]), tv_head(h[:, hidden:])

        self._target_heads = target_heads

        def target_hidden(x):
            return t_hidden_of(t_cnn(x))

        self._target_hidden = target_hidden
        self._thead_params = (ta_head.weight, ta_head.bias, tv_head.weight,
                              tv_head.bias)

    def _online_q(self, x):
        if getattr(self, "_fast_fwd", None) is not None:
            return self._fast_fwd(x)
        return self.net.forward([x])[0]

    # ------------------------------------------------------------------
    # ingest: transport -> pinned staging -> device replay (side stream)
    # ------------------------------------------------------------------
    _STAGE_ROWS = 4096  # persistent pinned staging capacity (C2)

    def _staging_buffers(self):
        if self._staging is None:
            from ..replay import make_apex_schema

            wire = make_apex_schema()  # wire format is always NCHW u8
            self._wire_names = list(wire)
            self._staging = {
                name: torch.empty((self._STAGE_ROWS, *shape), dtype=dtype
                                  ).pin_memory()
                for name, (shape, dtype) in wire.items()
            }
            self._staging["__prio__"] = torch.empty(
                self._STAGE_ROWS, dtype=torch.float32).pin_memory()
            self._stage_evt = torch.cuda.Event()
            self._stage_busy = False
        return self._staging

    def ingest(self) -> int:
        if self.transport is None:
            return 0
        if self._ingest_stream is not None and \
                hasattr(self.transport, "drain_views"):
            return self._ingest_views()
        got = self.transport.drain()
        if got is None:
            return 0
        cols_np, prio_np = got
        n = len(prio_np)
        if self._ingest_stream is None:
            cols = {k: torch.from_numpy(np.ascontiguousarray(v))
                    for k, v in cols_np.items()}
            self.replay.push(cols, torch.from_numpy(
                np.ascontiguousarray(prio_np)))
            self.ingested_total += n
            return n
        # pinned staging ring: one hipHostMalloc for the process lifetime,
        # chunked numpy->pinned memcpy + async H2D on the side stream
        stage = self._staging_buffers()
        done = 0
        while done < n:
            k = min(self._STAGE_ROWS, n - done)
            if self._stage_busy:
                # the pinned buffers may still be read by a previous async
                # H2D — wait_stream below only orders GPU streams, it does
                # NOT protect the host-side refill from racing that copy
                self._stage_evt.synchronize()
            for name, arr in cols_np.items():
                stage[name][:k].numpy()[:] = arr[done : done + k]
            stage["__prio__"][:k].numpy()[:] = prio_np[done : done + k]
            with torch.cuda.stream(self._ingest_stream):
                dev_cols = {
                    name: stage[name][:k].to(self.device, non_blocking=True)
                    for name in cols_np
                }
                if self._nhwc:
                    for kk in ("state", "next_state"):
                        dev_cols[kk] = dev_cols[kk].permute(0, 2, 3, 1
                                                            ).contiguous()
                if self.state_dtype != torch.uint8:
                    # fp16-compressed replay stores frames pre-scaled
                    for kk in ("state", "next_state"):
                        dev_cols[kk] = dev_cols[kk].to(self.state_dtype) / 255.0
                prio_dev = stage["__prio__"][:k].to(self.device,
                                                    non_blocking=True)
            # H2D copies + relayout overlap queued compute, but the ring/
            # sum-tree MUTATION must not interleave with the compute
            # stream's own tree ops (sample/priority rebuild): two
            # concurrent level-by-level rebuilds can leave the root
            # transiently inconsistent with the leaves, and a sample in
            # that window could descend into an unwritten row
            self._ingest_stream.wait_stream(
                torch.cuda.current_stream(self.device))
            with torch.cuda.stream(self._ingest_stream):
                self.replay.push(dev_cols, prio_dev)
            self._stage_evt.record(self._ingest_stream)
            self._stage_busy = True
            # replay state is consumed by the compute stream; order it after
            # the ingest stream
            torch.cuda.current_stream(self.device).wait_stream(
                self._ingest_stream)
            done += k
        self.ingested_total += n
        return n

    def _ingest_views(self) -> int:
        """Fast ingest: ONE host copy per byte — the actor rings' shm
        records (AoS) are column-copied straight into the pinned staging
        buffers (SoA), then hipMemcpyAsync to HBM on the side stream. The
        reference's equivalent path (Redis lrange + unpickle + np.stack,
        APE_X/ReplayMemory.py:118-146) copies every byte 4+ times."""
        stage = self._staging_buffers()
        np_dtype = self.transport.record_dtype
        names = list(self._wire_names)
        ext = ops.hip_ext()
        if getattr(self, "_pack_spec", None) is None:
            f = np_dtype.fields  # name -> (dtype, offset)
            keys = names + ["priority"]
            self._pack_spec = (
                [int(f[n][1]) for n in keys],
                [int(f[n][0].itemsize) for n in keys],
                [stage[n] for n in names] + [stage["__prio__"]],
                int(np_dtype.itemsize),
                ext is not None and hasattr(ext, "pack_rows"),
            )
        offs, sizes, dsts, rec_size, has_pack = self._pack_spec
        total = 0
        cursor = 0

        def flush():
            nonlocal cursor
            k = cursor
            if k == 0:
                return
            with torch.cuda.stream(self._ingest_stream):
                dev_cols = {
                    name: stage[name][:k].to(self.device, non_blocking=True)
                    for name in names
                }
                if self._nhwc:
                    for kk in ("state", "next_state"):
                        dev_cols[kk] = dev_cols[kk].permute(0, 2, 3, 1
                                                            ).contiguous()
                if self.state_dtype != torch.uint8:
                    for kk in ("state", "next_state"):
                        dev_cols[kk] = dev_cols[kk].to(self.state_dtype) / 255.0
                prio_dev = stage["__prio__"][:k].to(self.device,
                                                    non_blocking=True)
            # tree/ring mutation ordered after queued compute (see ingest)
            self._ingest_stream.wait_stream(
                torch.cuda.current_stream(self.device))
            with torch.cuda.stream(self._ingest_stream):
                self.replay.push(dev_cols, prio_dev)
            self._stage_evt.record(self._ingest_stream)
            self._stage_busy = True
            torch.cuda.current_stream(self.device).wait_stream(
                self._ingest_stream)
            cursor = 0

        stop_evt = self._ingest_stop
        for views, n, advance in self.transport.drain_views(
                max_per_ring=self._STAGE_ROWS):
            if stop_evt is not None and stop_evt.is_set():
                break  # prompt exit mid-sweep (thread shutdown)
            for v in views:
                nrows = v.shape[0]
                i = 0
                while i < nrows:
                    if cursor == 0 and self._stage_busy:
                        # previous async H2D may still read the pinned bufs
                        self._stage_evt.synchronize()
                        self._stage_busy = False
                    k = min(self._STAGE_ROWS - cursor, nrows - i)
                    if has_pack:
                        # GIL-free multithreaded AoS->SoA scatter (C++)
                        src = torch.from_numpy(
                            np.ascontiguousarray(v[i : i + k]).reshape(-1))
                        ext.pack_rows(src, rec_size, offs, sizes, dsts, cursor)
                    else:
                        sl = v[i : i + k].reshape(-1).view(np_dtype)
                        for name in names:
                            stage[name][cursor : cursor + k].numpy()[:] = \
                                sl[name]
                        stage["__prio__"][cursor : cursor + k].numpy()[:] = \
                            sl["priority"]
                    cursor += k
                    i += k
                    total += k
                    self.ingested_total += k  # incremental: telemetry must
                    # not wait for the whole sweep (a post-capture backlog
                    # once made one call span seconds, reading as "0 rows")
                    if cursor == self._STAGE_ROWS:
                        flush()
            advance(n)
        flush()
        return total

    def push_experience(self, cols: Dict[str, torch.Tensor], prio: torch.Tensor):
        """Direct (in-process) push; frames arrive NCHW uint8 (wire format)."""
        cols = dict(cols)
        for k in ("state", "next_state"):
            v = cols[k]
            if self._nhwc:
                v = v.permute(0, 2, 3, 1)
            if self.state_dtype != torch.uint8 and v.dtype == torch.uint8:
                v = v.to(self.state_dtype) / 255.0
            cols[k] = v.contiguous()
        self.replay.push(cols, prio)

    # ------------------------------------------------------------------
    # train
    # ------------------------------------------------------------------
    def train_step(self, data, idx, weights) -> Dict[str, torch.Tensor]:
        stats, prio = self._fwd_bwd(data, weights)
        if self.mp is not None:
            self.mp.allreduce_grads()
            self._optimize_mp()
        else:
            if self.reducer is not None:
                self.reducer.all_reduce()
            self.optim.step()
        self.replay.update(idx, prio)
        return stats

    def _fwd_bwd(self, data, weights):
        """Sample-batch forward + fused TD loss + backward; grads are left
        in the flat buffers (collective-free -> hipGraph-capturable).
        Returns (stats, prio)."""
        cuda = self.device.type == "cuda"
        if cuda:
            # NHWC frames straight into the fused conv stack (u8: dequant
            # fused into conv1; fp16-compressed: one cast to bf16). Batches
            # from a remote replay arrive NCHW on CPU -> move + relayout.
            def to_cl(t):
                if t.shape[-1] == 4:  # NHWC storage
                    return t.to(self.device, non_blocking=True).permute(0, 3, 1, 2)
                return t.to(self.device, non_blocking=True).contiguous(
                    memory_format=torch.channels_last
                )

            s = to_cl(data["state"])
            sp = to_cl(data["next_state"])
            if s.dtype == torch.float16:
                s = s.to(torch.bfloat16)
                sp = sp.to(torch.bfloat16)
        elif data["state"].dtype == torch.uint8:
            s = ops.dequant_frames(data["state"], torch.float32)
            sp = ops.dequant_frames(data["next_state"], torch.float32)
        else:
            s = data["state"].float()
            sp = data["next_state"].float()
        actions = data["action"].to(self.device).long()
        rewards = data["reward"].to(self.device)
        dones = data["done"].to(self.device)
        weights = weights.to(self.device)

        if cuda and getattr(self, "_use_q_loss", False):
            # deepest fusion: head projections + dueling + TD loss in ONE
            # kernel; backward = closed-form dh + one head-grad reduction
            h_s = self._fast_hidden(s)
            with torch.no_grad():
                h_on = self._fast_hidden(sp)
                h_tg = self._target_hidden(sp)
            wa, ba, wv, bv = self._head_params
            wat, bat, wvt, bvt = self._thead_params
            loss, prio, qmean = ops.dueling_q_head_loss(
                h_s, wa, ba, wv, bv, h_on, h_tg, wat, bat, wvt, bvt,
                actions, rewards, dones, weights, self.gamma, self.n_step,
                self.alpha,
            )
        elif cuda and getattr(self, "_fast_heads", None) is not None:
            # fused whole-head path: dueling epilogues live inside the loss
            # kernel; backward writes (g_adv, g_val) closed-form
            adv_s, val_s = self._fast_heads(s)
            with torch.no_grad():
                adv_on, val_on = self._fast_heads(sp)
                adv_tg, val_tg = self._target_heads(sp)
            loss, prio, qmean = ops.dueling_nstep_dqn_loss(
                adv_s, val_s, adv_on, val_on, adv_tg, val_tg, actions,
                rewards, dones, weights, self.gamma, self.n_step, self.alpha,
            )
        else:
            q_s = self._online_q(s) if cuda else self.net.forward([s])[0]
            with torch.no_grad():
                q_sp_on = (self._online_q(sp) if cuda
                           else self.net.forward([sp])[0])
                q_sp_tg = self.target.forward([sp])[0]
            loss, prio, qmean = ops.nstep_dqn_loss(
                q_s, q_sp_on, q_sp_tg, actions, rewards,
                dones, weights, self.gamma, self.n_step, self.alpha,
                with_value_stat=True,
            )
        if self.mp is not None:
            self.mp.direct_grads(loss, getattr(self, "_direct_list", None))
        else:
            self.optim.zero_grad(set_to_none=False)
            loss.backward()
        return {"loss": loss.detach(), "value": qmean}, prio

    def _optimize_mp(self):
        """Upcast + optimizer + param sync (post-collective stage;
        hipGraph-capturable)."""
        self.mp.upcast_grads()
        self.optim.step()
        self.mp.sync_compute_params()

    def sync_target(self):
        """Hard target sync (APE_X/Learner.py:204-208, tau=1)."""
        if self.mp is not None:
            self.flat_tparam.copy_(self.mp.flat_cparam)
        else:
            self.target.updateParameter(self.model, 1.0)

    def _inner_step(self) -> Dict[str, torch.Tensor]:
        data, idx, w = self.replay.sample(self.batch_size, self.beta)
        return self.train_step(data, idx, w)

    def _cadence(self):
        self.step_count += 1
        if self.step_count % self.cfg.target_frequency == 0:
            self.sync_target()
            self.publish_weights(include_target=True)
        elif self.step_count % PUBLISH_EVERY == 0:
            self.publish_weights()

    def step(self) -> Dict[str, torch.Tensor]:
        stats = self.maybe_profile_first_step(self._inner_step)
        self._cadence()
        return stats

    def make_graphed_step(self, warmup_iters: int = 3):
        """hipGraph-capture the learner step (PER sample -> fused-conv
        forwards -> fused loss -> backward -> grad cast -> optimizer ->
        priority update). All shapes are static and RNG lives in a device
        seed buffer, so one graph replay per learner step. Cadence ops
        (target sync, weight publish) stay eager. Requires a warm, fixed-size
        replay (n_valid is baked into the sample kernel).

        At world_size > 1 the step is captured as TWO graphs with the RCCL
        all-reduce running eagerly between them — collectives are never
        captured, so the multi-GPU path cannot be broken by graph-capture
        support gaps in the collective stack."""
        assert self.device.type == "cuda", "graph capture needs a GPU"
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
            self._graph = graph  # keep alive (owns the memory pool)

            def stepper():
                graph.replay()
                self._cadence()
                return static_out

            return stepper

        # ---- overlapped pipeline (north-star C1): 4 graphs
        #   g_s   sample batch t+1 into static buffers
        #   g_fb  fwd + fused loss + backward (+ save idx for the update)
        #   g_upd priority update of batch t
        #   g_opt upcast + optimizer + param sync
        # Step t runs: g_fb | [comm stream: RCCL all-reduce(flat_cgrad)]
        #              in parallel with [compute stream: g_upd, g_s] | join
        #              | g_opt — the collective is hidden behind the
        #              priority update + NEXT replay sample. Op order on the
        #              replay (update-then-sample) is identical to the
        #              sequential path, so trajectories are bit-equal.
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
            comm.wait_stream(cur)  # grads ready on the compute stream
            with torch.cuda.stream(comm):
                mp.allreduce_grads()  # RCCL on the comm stream
            g_upd.replay()  # overlap window: priority update of batch t
            g_s.replay()  # ... and sample of batch t+1
            cur.wait_stream(comm)
            g_opt.replay()
            self._cadence()
            return static_out

        return stepper

    def make_pipelined_step(self):
        """Eager pipelined step for world_size > 1 without graph capture
        (CPU/gloo rehearsal of the same overlap ordering; also the GPU
        fallback if capture fails). The async all-reduce runs while the
        priority update + next sample execute; op order on the replay is
        identical to the sequential stepper."""
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
                self.optim.step()
            self._cadence()
            return stats

        return stepper

    # ------------------------------------------------------------------
    # weight publication (seqlock bus; replaces Redis state_dict keys)
    # ------------------------------------------------------------------
    def publish_weights(self, include_target: bool = False):
        if self.transport is None or self.rank != 0:
            return
        if self.request_publish(include_target):
            return  # async publisher thread will snapshot + publish
        self._publish_sync(include_target, self.step_count)

    def _publish_sync(self, include_target: bool, count: int):
        cpu_sd = self.snapshot_state_dict()
        payload: Dict[str, Any] = {"count": count, "state_dict": cpu_sd}
        if include_target:
            payload["target_state_dict"] = {
                k: v.detach().to("cpu", torch.float32)
                for k, v in self.target.state_dict().items()
            }
        self.transport.publish(payload)

    # ------------------------------------------------------------------
    # run loop (telemetry cadence per SURVEY §5.5)
    # ------------------------------------------------------------------
    def wait_memory(self, min_items: Optional[int] = None, timeout: float = 600.0):
        need = min_items if min_items is not None else self.cfg.buffer_size
        t0 = time.time()
        while len(self.replay) <= need:
            if self._ingest_thread is None:
                with self._ingest_lock:
                    self.ingest()
            if time.time() - t0 > timeout:
                raise TimeoutError(
                    f"replay warmup stalled at {len(self.replay)}/{need}"
                )
            time.sleep(0.01)

    def run(self, max_steps: int = 1_000_000, warmup_items: Optional[int] = None):
        # ingest runs on its own daemon thread (reference parity: the Replay
        # drain thread) so drain+pin+H2D never stall the train loop
        self.start_ingest_thread()
        self.wait_memory(warmup_items)
        self.publish_weights(include_target=True)
        last_loss = None
        stepper = None  # hipGraph-captured once the replay ring is full
        while self.step_count < max_steps:
            t0 = time.perf_counter()
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
                    print(f"[APE_X] graph capture failed ({e}); staying eager",
                          flush=True)
                    stepper = self.step
                self.start_ingest_thread()
            stats = (stepper or self.step)()
            self.time_block("train", time.perf_counter() - t0)
            last_loss = stats["loss"]
            if self.step_count % self.LOG_EVERY == 0:
                self._log_block(stats)
            if self.step_count % self.CKPT_EVERY == 0:
                self.save_checkpoint()
        return last_loss

    def _log_block(self, stats):
        # eps rides the reward records: report both the fleet mean and the
        # near-greedy mean (the reference's Reward gates on eps < 0.05,
        # APE_X/Player.py:272-277). No fabricated -21 placeholder when
        # nothing was drained.
        pairs = (self.transport.drain_rewards_with_eps()
                 if self.transport is not None
                 and hasattr(self.transport, "drain_rewards_with_eps")
                 else [(r, 0.0) for r in (self.transport.drain_rewards()
                                          if self.transport else [])])
        rewards = [r for r, _ in pairs]
        greedy = [r for r, e in pairs if e < 0.05]
        mean_r = float(np.mean(rewards)) if rewards else float("nan")
        mean_g = float(np.mean(greedy)) if greedy else float("nan")
        timing = self.flush_timing()
        loss = float(stats["loss"])
        value = float(stats["value"])
        norm = float(self.model.calculateNorm())
        if greedy:
            self.log_scalar("Reward", mean_g)  # reference semantics
        if rewards:
            self.log_scalar("Reward_all", mean_r)
        self.log_scalar("value", value)
        self.log_scalar("norm", norm)
        self.log_scalar("loss", loss)
        if self.rank == 0:
            sps = self.LOG_EVERY / max(timing["wall"], 1e-9)
            print(
                f"[APE_X] step={self.step_count} loss={loss:.5f} value={value:.3f} "
                f"norm={norm:.2f} reward={mean_r:.1f} "
                f"greedy_reward={mean_g:.1f} replay={len(self.replay)} "
                f"steps/s={sps:.1f} "
                + " ".join(f"{k}={v:.2f}s" for k, v in timing.items()),
                flush=True,
            )

    # -- checkpoint -------------------------------------------------------
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
        if self.mp is not None:
            self.mp.sync_compute_params()
            self.sync_target()
        else:
            self.target.load_state_dict(sd)


# ===========================================================================
# Actor
# ===========================================================================


class LocalBuffer:
    """n-step transition assembly (APE_X/Player.py:19-60 semantics: emit
    [s_t, a_t, sum gamma^i r, s_{t+n}, done]); one transition per env step
    once warm, flushing the tail with done=1 at episode end.

    Deliberate divergence: the reference consumes n steps per emit
    (non-overlapping windows, APE_X/Player.py:56 ``del storage[:3n]``) and
    DROPS the non-aligned tail at episode end; we emit the Ape-X paper's
    per-step sliding windows (every (s_t, a_t) becomes a transition) and
    flush the tail with truncated returns. Terminal-step emissions carry
    done=1 like the reference's (Player.py:33-44)."""

    def __init__(self, n_step: int, gamma: float):
        self.n = n_step
        self.gamma = gamma
        self.buf: deque = deque()

    def append(self, state, action, reward):
        self.buf.append((state, action, reward))

    def emit_ready(self, next_state, done: bool) -> List[tuple]:
        """Called after each env step with s_{t+1}; returns finished
        transitions."""
        out = []
        if len(self.buf) >= self.n:
            s0, a0, _ = self.buf[0]
            r = 0.0
            for i in range(self.n):
                r += (self.gamma ** i) * self.buf[i][2]
            # a full window emitted AT the terminal step still ends the
            # episode: its bootstrap state is post-terminal, so done rides
            # the emission step's flag (APE_X/Player.py get_traj(done))
            out.append((s0, a0, r, next_state, 1.0 if done else 0.0))
            self.buf.popleft()
        if done:
            # flush remaining with truncated returns, done=1
            while self.buf:
                s0, a0, _ = self.buf[0]
                r = 0.0
                for i in range(len(self.buf)):
                    r += (self.gamma ** i) * self.buf[i][2]
                out.append((s0, a0, r, next_state, 1.0))
                self.buf.popleft()
        return out

    def clear(self):
        self.buf.clear()


class ApexPlayer:
    """CPU actor: env loop + eps-greedy inference + n-step assembly +
    actor-side priorities, pushing to the transport ring."""

    PUSH_BATCH = 16

    def __init__(self, cfg: Config, idx: int, transport, env=None,
                 env_kind: str = "auto", seed: Optional[int] = None,
                 max_staleness: Optional[int] = None):
        from ..actors.env import make_env

        self.cfg = cfg
        self.idx = idx
        self.transport = transport
        self.env = env or make_env(
            env_kind, seed=seed if seed is not None else idx,
            reward_clip=cfg.use_reward_clip,
        )
        self.device = torch.device(cfg.actor_device)
        # bounded staleness (SURVEY §5.3 gap): if the learner's published
        # `count` hasn't advanced for this many env steps, the actor blocks
        # instead of generating arbitrarily off-policy data. None = the
        # reference's behavior (proceed forever on stale weights).
        self.max_staleness = (
            max_staleness if max_staleness is not None
            else cfg.get("MAX_STALENESS_STEPS")
        )
        self._steps_since_fresh = 0
        self.model = BaseAgent(cfg.model_info).to(self.device).eval()
        self.target = BaseAgent(cfg.model_info).to(self.device).eval()
        n_actors = max(cfg.num_actors, 2)
        self.eps = 0.4 ** (1 + 7 * idx / (n_actors - 1))  # Player.py:78
        self.gamma = cfg.gamma
        self.n_step = cfg.unroll_step
        self.alpha = cfg.alpha
        self.action_n = cfg.action_size
        self.local = LocalBuffer(self.n_step, self.gamma)
        self.report_all_rewards = bool(cfg.get("REPORT_ALL_REWARDS", True))
        self.pending: List[tuple] = []
        self.env_steps = 0
        self.weight_version = -1
        self.rng = np.random.default_rng(1000 + idx)

    # -- inference -------------------------------------------------------
    @torch.no_grad()
    def act(self, state_u8: np.ndarray) -> int:
        if self.rng.random() < self.eps:
            return int(self.rng.integers(0, self.action_n))
        x = torch.from_numpy(state_u8).unsqueeze(0).float() / 255.0
        q = self.model.forward([x])[0]
        return int(q.argmax(1).item())

    # -- priorities ------------------------------------------------------
    @torch.no_grad()
    def _priorities(self, trans: List[tuple]) -> np.ndarray:
        """Double-DQN TD priority for fresh transitions (Player.py:135-159)."""
        s = torch.from_numpy(np.stack([t[0] for t in trans])).float() / 255.0
        sp = torch.from_numpy(np.stack([t[3] for t in trans])).float() / 255.0
        a = torch.tensor([t[1] for t in trans], dtype=torch.int64)
        r = torch.tensor([t[2] for t in trans], dtype=torch.float32)
        d = torch.tensor([t[4] for t in trans], dtype=torch.float32)
        q_s = self.model.forward([s])[0]
        q_sp_on = self.model.forward([sp])[0]
        q_sp_tg = self.target.forward([sp])[0]
        a_star = q_sp_on.argmax(1, keepdim=True)
        tgt = r + (self.gamma ** self.n_step) * q_sp_tg.gather(1, a_star).squeeze(1) * (1 - d)
        td = (tgt - q_s.gather(1, a.unsqueeze(1)).squeeze(1)).clamp(-1, 1)
        return ((td.abs() + 1e-7) ** self.alpha).numpy()

    def _flush(self):
        if not self.pending:
            return
        trans = self.pending
        self.pending = []
        self._push_trans(trans, self._priorities(trans))

    def _push_trans(self, trans: List[tuple], prio: np.ndarray):
        cols = {
            "state": np.stack([t[0] for t in trans]),
            "action": np.array([t[1] for t in trans], np.int32),
            "reward": np.array([t[2] for t in trans], np.float32),
            "next_state": np.stack([t[3] for t in trans]),
            "done": np.array([t[4] for t in trans], np.float32),
        }
        self.transport.push(cols, prio)

    # -- weights ---------------------------------------------------------
    def pull_weights(self):
        payload = self.transport.fetch()
        if payload is None:
            return
        count = payload.get("count", 0)
        if count == self.weight_version:
            return
        self.model.load_state_dict(payload["state_dict"])
        if "target_state_dict" in payload:
            self.target.load_state_dict(payload["target_state_dict"])
        self.weight_version = count
        self._steps_since_fresh = 0

    def _staleness_gate(self):
        if self.max_staleness is None:
            return
        self._steps_since_fresh += 1
        while self._steps_since_fresh > int(self.max_staleness):
            time.sleep(0.05)
            before = self.weight_version
            self.pull_weights()
            if self.weight_version != before:
                break

    # push via transport: ShmTransport actor-side adapter provides .push
    def run(self, max_env_steps: int = 1_000_000):
        self.pull_weights()
        episode_reward = 0.0
        state = self.env.reset()
        while self.env_steps < max_env_steps:
            action = self.act(state)
            next_state, reward, done, info = self.env.step(action)
            episode_reward += reward
            self.local.append(state, action, reward)
            finished = self.local.emit_ready(
                next_state, done or info.get("pseudo_done", False)
            )
            self.pending.extend(finished)
            if len(self.pending) >= self.PUSH_BATCH or done:
                self._flush()
            state = next_state
            self.env_steps += 1
            if self.env_steps % ACTOR_PULL_EVERY == 0:
                self.pull_weights()
            self._staleness_gate()
            if done:
                # reference gates reward telemetry to near-greedy actors
                # (Player.py:272-277); report_all_rewards widens it so the
                # learner's Reward scalar reflects the whole fleet
                if self.eps < 0.05 or self.report_all_rewards:
                    self.transport.push_reward(self.idx, episode_reward, self.eps)
                episode_reward = 0.0
                self.local.clear()
                state = self.env.reset()
        self._flush()


def run_apex_vec(players: List["ApexPlayer"], max_env_steps: int = 1_000_000):
    """Vectorized multi-env actor loop: M ApexPlayers in ONE process share
    the lead player's model/target, and every env step all M eps-greedy
    argmaxes come from a single batched forward instead of M single-frame
    forwards (the reference pins one Ray process per env,
    run_actor.py:46-55; at 256 actors that is host-process bound).

    Per-virtual-actor semantics are preserved exactly: each player keeps
    its own env, eps ladder slot, rng stream, n-step LocalBuffer,
    transport ring, and reward telemetry; only the weights and the
    argmax batch are shared. Weight pulls / the bounded-staleness gate
    run on the lead player (one shared model = one version for all M).

    Priority forwards are ALSO cross-player batched: players whose
    pending buffer is flush-ready after a step are flushed together with
    one _priorities pass over the concatenation (3 forwards of M*16
    instead of 3M forwards of 16 — the per-player flushes were the
    dominant serial cost in the first A/B, profiles/r02_vec_actor.md),
    then split back to each player's own ring."""
    assert players, "empty player list"
    lead = players[0]
    for p in players[1:]:
        p.model = lead.model
        p.target = lead.target
    lead.pull_weights()
    states = [p.env.reset() for p in players]
    ep_rew = [0.0] * len(players)

    def _batched_flush(ready: List["ApexPlayer"]):
        counts = []
        all_trans: List[tuple] = []
        for p in ready:
            counts.append(len(p.pending))
            all_trans.extend(p.pending)
            p.pending = []
        prio = lead._priorities(all_trans)
        off = 0
        for p, n in zip(ready, counts):
            p._push_trans(all_trans[off : off + n], prio[off : off + n])
            off += n

    while lead.env_steps < max_env_steps:
        x = torch.from_numpy(np.stack(states)).float().div_(255.0)
        with torch.no_grad():
            greedy = lead.model.forward([x])[0].argmax(1).tolist()
        flush_ready = []
        for j, p in enumerate(players):
            if p.rng.random() < p.eps:
                action = int(p.rng.integers(0, p.action_n))
            else:
                action = int(greedy[j])
            next_state, reward, done, info = p.env.step(action)
            ep_rew[j] += reward
            p.local.append(states[j], action, reward)
            p.pending.extend(p.local.emit_ready(
                next_state, done or info.get("pseudo_done", False)))
            if p.pending and (len(p.pending) >= p.PUSH_BATCH or done):
                flush_ready.append(p)
            states[j] = next_state
            p.env_steps += 1
            if done:
                if p.eps < 0.05 or p.report_all_rewards:
                    p.transport.push_reward(p.idx, ep_rew[j], p.eps)
                ep_rew[j] = 0.0
                p.local.clear()
                states[j] = p.env.reset()
        if flush_ready:
            _batched_flush(flush_ready)
        if lead.env_steps % ACTOR_PULL_EVERY == 0:
            lead.pull_weights()
        lead._staleness_gate()
    for p in players:
        p._flush()
