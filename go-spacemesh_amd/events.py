"""POST event bus — the subset of the reference's events package the hot
path emits and the verifier autoscaler consumes (events/events.go:34-64,
195-241; consumed at activation/post_verifier.go:55-71).

Events: InitStart, InitComplete, InitFailure (emitted by the init lifecycle,
activation/post.go:293-331) and PostStart, PostComplete (emitted around
proving, consumed by the autoscaler to free CPU/GPU for the prover)."""
from __future__ import annotations

import dataclasses
import threading
from typing import Callable, List


@dataclasses.dataclass(frozen=True)
class InitStart:
    node_id: bytes
    commitment_atx: bytes


@dataclasses.dataclass(frozen=True)
class InitComplete:
    node_id: bytes


@dataclasses.dataclass(frozen=True)
class InitFailure:
    node_id: bytes
    error: str


@dataclasses.dataclass(frozen=True)
class PostStart:
    node_id: bytes
    challenge: bytes


@dataclasses.dataclass(frozen=True)
class PostComplete:
    node_id: bytes


class EventBus:
    """Subscribe/emit with per-subscription filtering; delivery is
    synchronous in emit order (sufficient for the autoscaler contract)."""

    def __init__(self) -> None:
        self._mu = threading.Lock()
        self._subs: List[Callable[[object], None]] = []

    def subscribe(self, handler: Callable[[object], None]) -> Callable[[], None]:
        with self._mu:
            self._subs.append(handler)

        def unsubscribe() -> None:
            with self._mu:
                if handler in self._subs:
                    self._subs.remove(handler)
        return unsubscribe

    def emit(self, event: object) -> None:
        with self._mu:
            subs = list(self._subs)
        for h in subs:
            h(event)


_global_bus: EventBus | None = None
_global_mu = threading.Lock()


def bus() -> EventBus:
    global _global_bus
    with _global_mu:
        if _global_bus is None:
            _global_bus = EventBus()
        return _global_bus
