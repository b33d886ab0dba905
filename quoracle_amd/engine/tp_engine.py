"""Lockstep tensor-parallel engine: one LocalEngine replica per TP rank.

All logits-affecting state is replicated under TP (embeddings, norms,
residuals and logits are identical on every rank after the per-layer
all-reduces, and sampling uses per-request seeded generators), so the whole
continuous-batching engine runs as DETERMINISTIC LOCKSTEP REPLICAS: the
only synchronization is an admit-broadcast at the top of every engine step
— no per-token traffic.  The leader rank owns the real request queue and
result futures; replicas execute identical forwards so the per-layer RCCL
all-reduces line up (BASELINE config 5).

Group-level watchdog: a device fault on one rank defers its KV reset to
the next step's synchronization point, where a control-plane MAX
all-reduce of the pending-reset mask makes EVERY rank rebuild the same
models together — the lockstep invariant (identical block allocation,
identical batch composition) survives the reset.

Exactness requirements (hold by construction):
  * identical model shards from the same seed (models/llama.py TP sharding),
  * identical KV block allocation (deterministic free-list),
  * identical batch composition (this admit-broadcast),
  * bitwise-replicated logits (all-reduce output identical on all ranks) +
    per-request seeded generators => identical sampled tokens everywhere.
"""

from __future__ import annotations

import queue as _q
import sys
import time
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from ..parallel.tp import TPContext
from .engine import LocalEngine


class _StopSentinel:
    pass


class TPEngine(LocalEngine):
    def __init__(self, model_keys: Sequence[str], tp: TPContext,
                 device: Optional[torch.device] = None,
                 control_group=None, **kw):
        self.tp = tp
        self.control_group = control_group if control_group is not None \
            else tp.group
        self.is_leader = tp.rank == 0
        self.stop_seen = False
        self._idle_steps = 0
        kw.setdefault("tp", tp)
        super().__init__(model_keys, device=device, **kw)
        # lockstep: batches are composed ONLY from the admit-broadcast —
        # requests landing between broadcast and admit wait for the next
        # step (a leader admitting unannounced work would desync the
        # per-layer all-reduce counts and deadlock the group).  At
        # world=1 there is no broadcast, so the normal inbox admit must
        # stay on (found by the TP=1 GPU rehearsal: requests were never
        # admitted and generate_sync timed out empty).
        self._tp_direct = tp.world > 1
        # group watchdog: model keys pending a synchronized reset
        self._reset_pending: set = set()
        self._model_order = sorted(self.models)

    def request_stop(self) -> None:
        """Leader: broadcast shutdown to replicas on the next step."""
        if self.tp.world <= 1:
            # no broadcast at world 1 — the sentinel would land in the
            # NORMAL admit path, which expects sequences
            self.stop_seen = True
            return
        self._inbox.put(_StopSentinel())

    def reset_model(self, key: str) -> None:
        """Group-level watchdog: a rank-local fault must not reset one
        replica alone (block allocation would diverge and the per-layer
        all-reduces deadlock).  Defer to the next step's sync point where
        the whole group resets together."""
        if self.tp.world > 1:
            self._reset_pending.add(key)
            return
        super().reset_model(key)

    def _sync_group_resets(self) -> None:
        """MAX-all-reduce the pending-reset mask over the control group and
        apply the union on every rank at the same step boundary."""
        mask = torch.zeros(len(self._model_order), dtype=torch.int32)
        for i, key in enumerate(self._model_order):
            if key in self._reset_pending:
                mask[i] = 1
        dist.all_reduce(mask, op=dist.ReduceOp.MAX, group=self.control_group)
        self._reset_pending.clear()
        for i, key in enumerate(self._model_order):
            if int(mask[i]):
                super().reset_model(key)

    def step(self) -> bool:
        if self.tp.world > 1:
            # group watchdog first: apply any rank's pending resets
            # everywhere BEFORE admitting this step's requests, so fresh
            # work is never killed by a reset it arrived after
            self._sync_group_resets()
            if self.is_leader:
                drained = self._drain_inbox()
                stop = any(isinstance(s, _StopSentinel) for s in drained)
                seqs = [s for s in drained if not isinstance(s, _StopSentinel)]
                box = [([s.request for s in seqs], stop)]
                dist.broadcast_object_list(box, src=0,
                                           group=self.control_group)
                for s in seqs:
                    self._admit_seq(s)
                if stop:
                    self.stop_seen = True
            else:
                box = [None]
                dist.broadcast_object_list(box, src=0,
                                           group=self.control_group)
                reqs, stop = box[0]
                if stop:
                    self.stop_seen = True
                for req in reqs:
                    err = self._submit(req, None, enqueue=False)
                    if err is not None:
                        print(f"[tp-engine r{self.tp.rank}] divergent "
                              f"submit: {err.error}", file=sys.stderr,
                              flush=True)
                    else:
                        self._admit_seq(self._made_seq)
        worked = super().step()
        if self.tp.world > 1:
            # deterministic shared idle backoff: every rank sees the same
            # (worked, active) state, so all sleep together
            self._idle_steps = 0 if worked else self._idle_steps + 1
            if not worked and self._idle_steps > 3:
                time.sleep(0.004)
        return worked

    def _drain_inbox(self) -> List:
        drained = []
        while True:
            try:
                drained.append(self._inbox.get_nowait())
            except _q.Empty:
                return drained

    def _loop(self) -> None:
        # lockstep replicas can never block on the wake event — every rank
        # must reach the admit-broadcast each iteration
        while self._running and not self.stop_seen:
            self.step()

    def _finish(self, hm, done) -> None:
        if not self.is_leader:
            for seq in done:
                if seq in hm.active:
                    hm.active.remove(seq)
            return
        super()._finish(hm, done)


    def stop(self) -> None:
        if self.tp.world > 1 and self.is_leader and not self.stop_seen:
            self.request_stop()
            if self._thread is None:
                self.step()
            else:
                # the loop thread must broadcast the stop sentinel BEFORE
                # _running flips false, or replicas hang in their next
                # admit-broadcast against a dead leader
                deadline = time.monotonic() + 30
                while not self.stop_seen and time.monotonic() < deadline:
                    time.sleep(0.002)
        super().stop()


def serve_tp_replica(engine: TPEngine) -> None:
    """Replica main loop: lockstep-step until the leader broadcasts stop."""
    while not engine.stop_seen:
        engine.step()
