"""Pod schedule state machine (reference pkg/internal/types.go:154-198).

States and legal transitions:
  Waiting    -> Preempting | Binding | (deleted)
  Preempting -> Binding | Waiting | (deleted)
  Binding    -> Bound | (deleted)
  Bound      -> (deleted)
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

POD_WAITING = "Waiting"
POD_PREEMPTING = "Preempting"
POD_BINDING = "Binding"
POD_BOUND = "Bound"

_LEGAL_TRANSITIONS = {
    POD_WAITING: {POD_PREEMPTING, POD_BINDING, POD_WAITING},
    POD_PREEMPTING: {POD_BINDING, POD_WAITING, POD_PREEMPTING},
    POD_BINDING: {POD_BINDING, POD_BOUND},
    POD_BOUND: {POD_BOUND},
}


@dataclass
class PodScheduleStatus:
    pod: Dict[str, Any]
    state: str = POD_WAITING
    # number of bind attempts; force-bind kicks in at forcePodBindThreshold
    pod_bind_attempts: int = 0
    pod_scheduling_spec: Optional[Any] = None
    pod_bind_info: Optional[Any] = None
    # decided node while in Binding state
    node: str = ""
    # victims while in Preempting state
    victim_pod_keys: List[str] = field(default_factory=list)

    def transition(self, new_state: str) -> None:
        if new_state not in _LEGAL_TRANSITIONS.get(self.state, set()):
            raise RuntimeError(f"illegal pod state transition {self.state} -> {new_state}")
        self.state = new_state
