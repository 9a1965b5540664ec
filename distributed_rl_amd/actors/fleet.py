"""Actor fleet: one OS process per actor over a shared-memory transport
session — the Ray + Redis replacement (run_actor.py:46-55 spawned Ray
remote actors pinned to 1 CPU; here plain multiprocessing with the SPSC
ring transport)."""

from __future__ import annotations

import multiprocessing as mp
import os
import time
from typing import List, Optional

from ..algos import get_player_cls, get_vec_runner, get_wire_schema
from ..config import load_config
from .transport import ActorEndpoint, RecordCodec, TransportSession


def _actor_main(cfg_spec: str, idx: int, transport_dir: str,
                max_env_steps: int, env_kind: str, tcp: str = ""):
    # actors are CPU-only: keep torch single-threaded per actor
    os.environ.setdefault("OMP_NUM_THREADS", "1")
    import torch

    torch.set_num_threads(1)
    cfg = load_config(cfg_spec)
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    session = None
    if tcp:
        from .tcp_transport import TcpActorEndpoint

        host, port = tcp.rsplit(":", 1)
        endpoint = TcpActorEndpoint(host, int(port), codec, idx=idx)
    else:
        session = TransportSession(transport_dir, codec, num_rings=0,
                                   create=False)
        endpoint = ActorEndpoint(session, idx)
    player = get_player_cls(cfg.alg)(cfg, idx=idx, transport=endpoint,
                                     env_kind=env_kind)
    try:
        player.run(max_env_steps=max_env_steps)
    except KeyboardInterrupt:
        pass
    finally:
        if session is not None:
            session.close()


def _vec_actor_main(cfg_spec: str, indices: List[int], transport_dir: str,
                    max_env_steps: int, env_kind: str, tcp: str = ""):
    """M virtual actors in one process with a shared model and a batched
    per-step forward (see algos.ape_x.run_apex_vec). 2 intra-op threads:
    the batched conv forward scales past 1 thread, unlike the
    single-frame path."""
    os.environ.setdefault("OMP_NUM_THREADS", "2")
    import torch

    torch.set_num_threads(2)
    cfg = load_config(cfg_spec)
    runner = get_vec_runner(cfg.alg)
    if runner is None:
        raise ValueError(f"{cfg.alg} has no vectorized actor loop; "
                         "use envs_per_proc=1")
    schema, with_prio = get_wire_schema(cfg)
    codec = RecordCodec(schema, with_priority=with_prio)
    session = None
    if tcp:
        from .tcp_transport import TcpActorEndpoint

        host, port = tcp.rsplit(":", 1)
        endpoints = [TcpActorEndpoint(host, int(port), codec, idx=i)
                     for i in indices]
    else:
        session = TransportSession(transport_dir, codec, num_rings=0,
                                   create=False)
        endpoints = [ActorEndpoint(session, i) for i in indices]
    player_cls = get_player_cls(cfg.alg)
    players = [player_cls(cfg, idx=i, transport=ep, env_kind=env_kind)
               for i, ep in zip(indices, endpoints)]
    try:
        runner(players, max_env_steps=max_env_steps)
    except KeyboardInterrupt:
        pass
    finally:
        if session is not None:
            session.close()


class ActorFleet:
    """Spawn/supervise N actor processes. respawn_on_exit keeps the fleet at
    full strength (failure handling the reference lacks)."""

    def __init__(self, cfg_spec: str, num_actors: int, transport_dir: str,
                 start_idx: int = 0, env_kind: str = "auto",
                 max_env_steps: int = 1 << 60, respawn_on_exit: bool = True,
                 tcp: str = "", envs_per_proc: int = 1):
        self.cfg_spec = cfg_spec
        self.num_actors = num_actors
        self.start_idx = start_idx
        self.transport_dir = transport_dir
        self.tcp = tcp
        self.env_kind = env_kind
        self.max_env_steps = max_env_steps
        self.respawn = respawn_on_exit
        self.envs_per_proc = max(1, envs_per_proc)
        if self.envs_per_proc > 1:
            # fail at construction, not inside a respawn loop
            alg = load_config(cfg_spec).alg
            if get_vec_runner(alg) is None:
                raise ValueError(f"{alg} has no vectorized actor loop; "
                                 "use envs_per_proc=1")
        self.num_procs = -(-num_actors // self.envs_per_proc)
        self.ctx = mp.get_context("spawn")
        self.procs: List[Optional[mp.Process]] = [None] * self.num_procs

    def _slot_indices(self, slot: int) -> List[int]:
        lo = self.start_idx + slot * self.envs_per_proc
        hi = min(lo + self.envs_per_proc, self.start_idx + self.num_actors)
        return list(range(lo, hi))

    def _spawn(self, slot: int):
        indices = self._slot_indices(slot)
        if self.envs_per_proc == 1:
            target, who = _actor_main, indices[0]
        else:
            target, who = _vec_actor_main, indices
        p = self.ctx.Process(
            target=target,
            args=(self.cfg_spec, who, self.transport_dir, self.max_env_steps,
                  self.env_kind, self.tcp),
            daemon=True,
            name=f"drl-actor-{indices[0]}",
        )
        p.start()
        self.procs[slot] = p

    def start(self):
        for i in range(self.num_procs):
            self._spawn(i)

    def supervise(self, poll_s: float = 5.0):
        """Blocking loop: respawn dead actors (heartbeat = process liveness)."""
        while True:
            time.sleep(poll_s)
            alive = 0
            for i, p in enumerate(self.procs):
                if p is None or not p.is_alive():
                    if self.respawn:
                        self._spawn(i)
                else:
                    alive += 1
            if alive == 0 and not self.respawn:
                return

    def alive_count(self) -> int:
        return sum(1 for p in self.procs if p is not None and p.is_alive())

    def join(self, timeout: Optional[float] = None):
        for p in self.procs:
            if p is not None:
                p.join(timeout)

    def stop(self):
        for p in self.procs:
            if p is not None and p.is_alive():
                p.terminate()
        for p in self.procs:
            if p is not None:
                p.join(5)
