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

import fnmatch
import threading
from collections import defaultdict
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

    @staticmethod
    def _match(pattern: str, topic: str) -> bool:
        # MQTT wildcards: '+' one level, '#' remaining levels.
        glob = pattern.replace("+", "*").replace("#", "*")
        if pattern == topic:
            return True
        if "#" in pattern:
            prefix = pattern.split("#", 1)[0]
            return topic.startswith(prefix) or fnmatch.fnmatch(topic, glob)
        pp, tp = pattern.split("/"), topic.split("/")
        if len(pp) != len(tp):
            return False
        return all(p == "+" or p == t for p, t in zip(pp, tp))

    def publish(self, topic: str, payload: Any, retain: bool = False) -> None:
        with self._lock:
            if retain:
                self._retained[topic] = payload
            self._log.append((topic, payload))
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
