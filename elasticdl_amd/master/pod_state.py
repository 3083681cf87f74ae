"""Declarative pod state machine (reference: master/pod_state.py:28-118).

Transitions keyed by (from_status, event_type, phase); each carries a
``should_relaunch`` flag consumed by the PodManager.
"""

from dataclasses import dataclass
from typing import Optional

from elasticdl_amd.common.constants import PodStatus


class PodEventType:
    ADDED = "ADDED"
    MODIFIED = "MODIFIED"
    DELETED = "DELETED"


@dataclass(frozen=True)
class Transition:
    from_status: str
    event_type: str
    phase: Optional[str]  # None = any
    to_status: str
    should_relaunch: bool = False


POD_STATE_FLOW = [
    Transition(PodStatus.INITIAL, PodEventType.ADDED, "Pending", PodStatus.PENDING),
    Transition(PodStatus.INITIAL, PodEventType.ADDED, "Running", PodStatus.RUNNING),
    Transition(PodStatus.INITIAL, PodEventType.MODIFIED, "Pending", PodStatus.PENDING),
    # a watch (re)connect can surface a pod first via MODIFIED/DELETED at
    # any phase — INITIAL must accept them or the pod is tracked forever
    # as INITIAL and never relaunched
    Transition(PodStatus.INITIAL, PodEventType.MODIFIED, "Running", PodStatus.RUNNING),
    Transition(PodStatus.INITIAL, PodEventType.MODIFIED, "Succeeded", PodStatus.SUCCEEDED),
    Transition(PodStatus.INITIAL, PodEventType.MODIFIED, "Failed", PodStatus.FAILED, True),
    Transition(PodStatus.INITIAL, PodEventType.ADDED, "Succeeded", PodStatus.SUCCEEDED),
    Transition(PodStatus.INITIAL, PodEventType.ADDED, "Failed", PodStatus.FAILED, True),
    Transition(PodStatus.INITIAL, PodEventType.DELETED, None, PodStatus.DELETED, True),
    Transition(PodStatus.PENDING, PodEventType.MODIFIED, "Running", PodStatus.RUNNING),
    Transition(PodStatus.PENDING, PodEventType.MODIFIED, "Succeeded", PodStatus.SUCCEEDED),
    Transition(PodStatus.PENDING, PodEventType.MODIFIED, "Failed", PodStatus.FAILED, True),
    Transition(PodStatus.PENDING, PodEventType.DELETED, None, PodStatus.DELETED, True),
    Transition(PodStatus.RUNNING, PodEventType.MODIFIED, "Succeeded", PodStatus.SUCCEEDED),
    Transition(PodStatus.RUNNING, PodEventType.MODIFIED, "Failed", PodStatus.FAILED, True),
    Transition(PodStatus.RUNNING, PodEventType.DELETED, None, PodStatus.DELETED, True),
    Transition(PodStatus.SUCCEEDED, PodEventType.DELETED, None, PodStatus.DELETED),
    Transition(PodStatus.FAILED, PodEventType.DELETED, None, PodStatus.DELETED),
]


def get_transition(from_status: str, event_type: str,
                   phase: Optional[str]) -> Optional[Transition]:
    for t in POD_STATE_FLOW:
        if (
            t.from_status == from_status
            and t.event_type == event_type
            and (t.phase is None or t.phase == phase)
        ):
            return t
    return None
