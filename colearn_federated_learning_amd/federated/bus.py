"""In-process pub/sub message bus with MQTT-style semantics.

CoLearn's distinctive control plane is MQTT pub/sub: participants announce
availability on topics, the coordinator discovers them, assigns
trainer/evaluator roles, and broadcasts session start/stop (SURVEY.md §1a,
§2a row 3).  On a single 8-GPU node a broker would add nothing but latency,
so this bus preserves the *semantics* — topic strings, retained messages,
wildcard subscriptions, availability / role / session lifecycle — as an
in-process object.  The coordinator and clients only ever talk through it,
so a networked MQTT transport could be slotted in behind the same API.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Tuple


class Topics:
    """Topic layout mirroring CoLearn's MQTT control plane."""

    AVAILABILITY = "colearn/availability"          # client -> server
    ROLE = "colearn/role/{client_id}"              # server -> client
    SESSION = "colearn/session"                    # server broadcast
    ROUND = "colearn/round"                        # server broadcast
    RESULT = "colearn/result/{client_id}"          # client -> server
    METRICS = "colearn/metrics/{client_id}"        # evaluator -> server


@dataclass
class _Subscription:
    pattern: str
    callback: Callable[[str, Any], None]


@dataclass
class MessageBus:
    """Thread-safe topic pub/sub with MQTT-style ``+``/``#`` wildcards."""

    _subs: List[_Subscription] = field(default_factory=list)
    _retained: Dict[str, Any] = field(default_factory=dict)
    _lock: threading.RLock = field(default_factory=threading.RLock)
    _log: List[Tuple[str, Any]] = field(default_factory=list)
    # locally-originated messages pending cross-rank relay (BusRelay);
    # relayed messages are republished with relayed=True and never re-enter
    _outbox: List[Tuple[str, Any, bool]] = field(default_factory=list)

    @staticmethod
    def _match(pattern: str, topic: str) -> bool:
        # MQTT wildcards, level-wise: '+' matches exactly ONE level,
        # '#' (last level only) matches all remaining levels.
        if pattern == topic:
            return True
        pp, tp = pattern.split("/"), topic.split("/")
        if pp and pp[-1] == "#":
            pp = pp[:-1]
            if len(tp) < len(pp):
                return False
            tp = tp[: len(pp)]
        elif len(pp) != len(tp):
            return False
        return all(p == "+" or p == t for p, t in zip(pp, tp))

    def publish(self, topic: str, payload: Any, retain: bool = False,
                relayed: bool = False) -> None:
        with self._lock:
            if retain:
                self._retained[topic] = payload
            self._log.append((topic, payload))
            if not relayed:
                self._outbox.append((topic, payload, retain))
            cbs = [s.callback for s in self._subs if self._match(s.pattern, topic)]
        for cb in cbs:
            cb(topic, payload)

    def subscribe(self, pattern: str,
                  callback: Callable[[str, Any], None]) -> None:
        with self._lock:
            self._subs.append(_Subscription(pattern, callback))
            retained = [
                (t, p) for t, p in self._retained.items()
                if self._match(pattern, t)
            ]
        for t, p in retained:
            callback(t, p)

    def unsubscribe(self, pattern: str) -> None:
        with self._lock:
            self._subs = [s for s in self._subs if s.pattern != pattern]

    def collect(self, pattern: str) -> Dict[str, Any]:
        """Snapshot of retained messages matching a pattern."""
        with self._lock:
            return {t: p for t, p in self._retained.items()
                    if self._match(pattern, t)}


class BusRelay:
    """Cross-rank control plane: relays locally-originated bus traffic over
    the torch.distributed process group at round boundaries.

    CoLearn's MQTT broker makes every participant's availability / result /
    metrics messages visible to the coordinator regardless of which device
    hosts the client.  With clients sharded across
    RCCL ranks, each rank's in-process bus only sees its own clients;
    ``sync()`` is a collective ``all_gather_object`` of each rank's pending
    outbox, republished everywhere with ``relayed=True`` — so rank 0's
    registry, logs and retained-message snapshots reflect the whole node,
    exactly as a broker would.  Payloads are tiny control dicts (no model
    bytes), so the object collective's pickle cost is irrelevant.

    ``sync()`` must be called at the same points on every rank (it is a
    collective); the coordinator calls it after client announcement and at
    the end of every round.
    """

    def __init__(self, bus: MessageBus) -> None:
        self.bus = bus
        self._synced = 0  # outbox prefix already relayed
        self._group = None

    def _gloo_group(self):
        """Dedicated gloo subgroup for the control plane: object
        collectives stay on host (no GPU round trip through RCCL for
        pickled dicts), and control messages can't interleave with the
        data plane's in-flight RCCL collectives."""
        import torch.distributed as dist

        if self._group is None:
            if dist.get_backend() == "gloo":
                self._group = dist.group.WORLD
            else:
                self._group = dist.new_group(backend="gloo")
        return self._group

    def sync(self) -> None:
        import torch.distributed as dist

        if not dist.is_initialized() or dist.get_world_size() == 1:
            with self.bus._lock:
                self._synced = len(self.bus._outbox)
            return
        with self.bus._lock:
            pending = self.bus._outbox[self._synced:]
            self._synced = len(self.bus._outbox)
        gathered: List[Any] = [None] * dist.get_world_size()
        dist.all_gather_object(gathered, pending, group=self._gloo_group())
        rank = dist.get_rank()
        for r, msgs in enumerate(gathered):
            if r == rank or not msgs:
                continue
            for topic, payload, retain in msgs:
                self.bus.publish(topic, payload, retain=retain, relayed=True)
