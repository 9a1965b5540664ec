"""Shared learner scaffolding: device/dtype handling, telemetry cadence,
checkpoint/resume, weight publishing.

Telemetry parity: TB scalars + console block every 500
learner steps (APE_X/Learner.py:219-262), checkpoint ``weight.pth`` under
``weight/<ALG>/<run>/`` (APE_X/Learner.py:256-262). Unlike the reference
(write-only checkpoints) we also persist optimizer + step for
resume.
"""

from __future__ import annotations

import os
import time
from typing import Any, Dict, Optional

import torch

from ..config import Config
from ..models import BaseAgent, get_optim


class LearnerBase:
    LOG_EVERY = 500
    CKPT_EVERY = 500

    def __init__(self, cfg: Config, device: Optional[str] = None, rank: int = 0,
                 world_size: int = 1, run_root: str = ".",
                 run_name: Optional[str] = None, enable_tb: bool = True):
        self.cfg = cfg
        self.rank = rank
        self.world_size = world_size
        if device is None:
            device = cfg.learner_device if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.autocast_dtype = (
            torch.bfloat16 if self.device.type == "cuda" else torch.float32
        )
        self.step_count = 0
        self.run_name = run_name or cfg.run_name()
        self.run_root = run_root
        self._writer = None
        self._enable_tb = enable_tb and rank == 0
        self._timing: Dict[str, float] = {}
        self._t_block = time.perf_counter()
        # opt-in replay persistence (SURVEY §5.4 "PER state optional"):
        # off by default — a full Ape-X buffer is multi-GB on disk
        self.checkpoint_replay = os.environ.get("DRL_CKPT_REPLAY", "0") == "1"
        self.ingested_total = 0
        self._ingest_thread = None
        self._ingest_stop = None
        self._pub_thread = None
        import threading as _threading

        # serializes ingest() across the daemon thread and any inline
        # caller — a stop-join timeout must never leave two drains racing
        # on the shared staging buffers
        self._ingest_lock = _threading.Lock()

    # -- background ingest -------------------------------------------------
    def start_ingest_thread(self):
        """Run ingest() on a daemon thread (the reference's Replay daemon
        thread, APE_X/ReplayMemory.py:19-27, re-created): drain + host
        copies + async H2D happen concurrently with the learner loop; the
        numpy memcpys release the GIL, and stream-ordering events keep the
        replay mutation ordered against the compute stream."""
        import threading

        if self._ingest_thread is not None or self.transport is None:
            return
        self._ingest_stop = threading.Event()

        stop = self._ingest_stop

        def loop():
            while not stop.is_set():
                try:
                    with self._ingest_lock:
                        n = self.ingest()
                except Exception as e:  # pragma: no cover
                    print(f"[ingest-thread] died: {e!r}", flush=True)
                    return
                if n == 0:
                    time.sleep(0.002)

        self._ingest_thread = threading.Thread(target=loop, daemon=True,
                                               name="drl-ingest")
        self._ingest_thread.start()
        self.start_publisher_thread()

    def stop_ingest_thread(self):
        self.stop_publisher_thread()
        if self._ingest_thread is None:
            return
        self._ingest_stop.set()
        self._ingest_thread.join(60)
        if self._ingest_thread.is_alive():  # pragma: no cover
            print("[ingest-thread] did not stop within 60s (continuing; "
                  "the ingest lock keeps a restart safe)", flush=True)
        self._ingest_thread = None

    # -- async weight publisher -------------------------------------------
    def start_publisher_thread(self):
        """Publish weights from a daemon thread: the D2H snapshot copy is
        stream-ordered on the compute stream (consistent between step
        graphs) but the event wait + state-dict build + pickle + shm write
        block only the publisher — the reference pays its pickle+Redis cost
        inline in the train loop (APE_X/Learner.py:212-216; IMPALA
        publishes EVERY step, Learner.py:286-287)."""
        import threading

        if self._pub_thread is not None or self.transport is None \
                or self.rank != 0:
            return
        self._pub_req = None
        self._pub_stop = threading.Event()
        self._pub_wake = threading.Event()

        def loop():
            while not self._pub_stop.is_set():
                self._pub_wake.wait(0.2)
                self._pub_wake.clear()
                req = self._pub_req
                if req is None:
                    continue
                self._pub_req = None
                try:
                    self._publish_sync(include_target=req[1], count=req[0])
                except Exception as e:  # pragma: no cover
                    print(f"[publisher-thread] died: {e!r}", flush=True)
                    return

        self._pub_thread = threading.Thread(target=loop, daemon=True,
                                            name="drl-publish")
        self._pub_thread.start()

    def stop_publisher_thread(self):
        if getattr(self, "_pub_thread", None) is None:
            return
        self._pub_stop.set()
        self._pub_wake.set()
        self._pub_thread.join(10)
        self._pub_thread = None

    def request_publish(self, include_target: bool = False) -> bool:
        """Queue an async publish (coalescing: only the latest request
        survives). Returns False if the publisher thread is not running —
        caller should publish synchronously."""
        if getattr(self, "_pub_thread", None) is None:
            return False
        # never drop an include_target request in favor of a plain one
        prev = self._pub_req
        inc = include_target or (prev is not None and prev[1])
        self._pub_req = (self.step_count, inc)
        self._pub_wake.set()
        return True

    # -- model helpers ----------------------------------------------------
    def build_model(self) -> BaseAgent:
        return BaseAgent(self.cfg.model_info).to(self.device)

    def build_optim(self, model):
        if getattr(self, "mp", None) is not None:
            # fused flat-buffer optimizer over the mp master (K12)
            from ..parallel.flat_optim import make_flat_optimizer

            fo = make_flat_optimizer(self.cfg.optim_info, self.mp)
            if fo is not None:
                return fo
        return get_optim(self.cfg.optim_info, model)

    # -- telemetry --------------------------------------------------------
    @property
    def writer(self):
        if self._writer is None and self._enable_tb:
            try:
                from torch.utils.tensorboard import SummaryWriter
            except ImportError:
                # tensorboard is optional (absent from some images);
                # console telemetry keeps working
                print("[telemetry] tensorboard not importable; "
                      "TB scalars disabled", flush=True)
                self._enable_tb = False
                return None
            path = self.cfg.log_dir(self.run_root, self.run_name)
            os.makedirs(path, exist_ok=True)
            self._writer = SummaryWriter(path)
            from ..utils import writeTrainInfo

            self._writer.add_text("configuration", writeTrainInfo(self.cfg.raw).info, 0)
        return self._writer

    def maybe_profile_first_step(self, fn, *args, **kw):
        """cProfile the first train call (reference parity:
        APE_X/Learner.py:177-180 profiles iteration 0). Enabled by
        DRL_PROFILE_FIRST_STEP=1; prints cumulative-time top-20."""
        if self.step_count == 0 and os.environ.get("DRL_PROFILE_FIRST_STEP") == "1":
            import cProfile
            import pstats

            prof = cProfile.Profile()
            out = prof.runcall(fn, *args, **kw)
            pstats.Stats(prof).sort_stats("cumulative").print_stats(20)
            return out
        return fn(*args, **kw)

    def log_scalar(self, tag: str, value: float, step: Optional[int] = None):
        w = self.writer
        if w is not None:
            w.add_scalar(tag, value, self.step_count if step is None else step)

    def time_block(self, key: str, dt: float):
        self._timing[key] = self._timing.get(key, 0.0) + dt

    def flush_timing(self) -> Dict[str, float]:
        out = dict(self._timing)
        out["wall"] = time.perf_counter() - self._t_block
        self._timing = {}
        self._t_block = time.perf_counter()
        return out

    # -- weight snapshot ---------------------------------------------------
    def snapshot_state_dict(self):
        """CPU fp32 state_dict of the online model for transport publish.

        On GPU with the mixed-precision trainer this is ONE pinned D2H copy
        per dtype group off the flat fp32 master buffer + host-side clones,
        instead of ~30 per-tensor synchronous ``.to("cpu")`` copies (the
        reference pays the per-tensor cost every 50 steps,
        APE_X/Learner.py:264-272; at 1k+ steps/s the publish cadence is on
        the hot path)."""
        mp = getattr(self, "mp", None)
        if (mp is None or self.device.type != "cuda"
                or len(list(self.model.buffers())) > 0):
            return {k: v.detach().to("cpu", torch.float32)
                    for k, v in self.model.state_dict().items()}
        if getattr(self, "_pub_plan", None) is None:
            named = dict(self.model.named_parameters())
            ident = {id(p): n for n, p in named.items()}
            pins, plan = [], []
            for gi, g in enumerate(mp.groups):
                pins.append(torch.empty(g.flat_mparam.numel(),
                                        dtype=torch.float32).pin_memory())
                off = 0
                for cp, mparam in zip(g.c_params, g.m_params):
                    plan.append((ident[id(mparam)], gi, off, mparam))
                    off += cp.numel()
            self._pub_plan = (pins, plan, torch.cuda.Event())
        pins, plan, evt = self._pub_plan
        for gi, g in enumerate(mp.groups):
            pins[gi].copy_(g.flat_mparam, non_blocking=True)
        evt.record()
        evt.synchronize()
        from ..parallel.precision import _view_like

        return {
            name: _view_like(pins[gi][off : off + p.numel()], p)
            .clone(memory_format=torch.contiguous_format)
            for name, gi, off, p in plan
        }

    # -- checkpoint / resume ---------------------------------------------
    def checkpoint_dir(self) -> str:
        d = self.cfg.weight_dir(self.run_root, self.run_name)
        os.makedirs(d, exist_ok=True)
        return d

    def state_for_checkpoint(self) -> Dict[str, Any]:
        raise NotImplementedError

    def load_from_checkpoint(self, state: Dict[str, Any]) -> None:
        raise NotImplementedError

    def save_checkpoint(self) -> str:
        if self.rank != 0:
            return ""
        d = self.checkpoint_dir()
        state = self.state_for_checkpoint()
        if self.checkpoint_replay and getattr(self, "replay", None) is not None:
            state["replay"] = self.replay.state_dict()
        # reference-format model-only file (contract: weight.pth)
        torch.save(state["model"], os.path.join(d, "weight.pth"))
        torch.save(state, os.path.join(d, "resume.pt"))
        return os.path.join(d, "weight.pth")

    def resume(self, path: str) -> None:
        """Load either a reference-style weight.pth (model only) or a full
        resume.pt (model + optimizer + step counter + optional replay)."""
        state = torch.load(path, map_location=self.device, weights_only=False)
        if isinstance(state, dict) and "model" in state and "step" in state:
            self.load_from_checkpoint(state)
            if "replay" in state and getattr(self, "replay", None) is not None:
                self.replay.load_state_dict(state["replay"])
        else:
            self.load_model_only(state)

    def load_model_only(self, sd) -> None:
        raise NotImplementedError
