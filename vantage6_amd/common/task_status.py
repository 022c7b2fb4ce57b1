"""Task/Run status taxonomy.

Reference parity: ``vantage6-common/vantage6/common/task_status.py`` —
``TaskStatus``/``RunStatus`` enum and the ``has_task_finished`` /
``has_task_failed`` helpers. The status strings are part of
the client-facing contract (they appear verbatim in REST responses).
"""
from enum import Enum


class RunStatus(str, Enum):
    # non-terminal
    PENDING = "pending"
    INITIALIZING = "initializing"
    ACTIVE = "active"
    # terminal — success
    COMPLETED = "completed"
    # terminal — failure taxonomy
    CRASHED = "crashed"
    KILLED = "killed by user"
    FAILED = "failed"
    START_FAILED = "start failed"
    NOT_ALLOWED = "not allowed"
    NO_IMAGE = "non-existing algorithm"

    def __str__(self) -> str:  # str(TaskStatus.PENDING) == "pending"
        return self.value


# vantage6 uses the same enum for tasks and runs (a task's status is derived
# from its runs); keep the alias for API parity.
TaskStatus = RunStatus

_FAILED = {
    RunStatus.CRASHED,
    RunStatus.KILLED,
    RunStatus.FAILED,
    RunStatus.START_FAILED,
    RunStatus.NOT_ALLOWED,
    RunStatus.NO_IMAGE,
}


def has_task_failed(status) -> bool:
    return RunStatus(status) in _FAILED


def has_task_finished(status) -> bool:
    return RunStatus(status) in _FAILED or RunStatus(status) == RunStatus.COMPLETED


def task_status_from_runs(run_statuses) -> RunStatus:
    """Derive a task-level status from its runs (server-side aggregation)."""
    statuses = [RunStatus(s) for s in run_statuses]
    if not statuses:
        return RunStatus.PENDING
    for s in statuses:
        if has_task_failed(s):
            return s
    if all(s == RunStatus.COMPLETED for s in statuses):
        return RunStatus.COMPLETED
    if any(s == RunStatus.ACTIVE for s in statuses):
        return RunStatus.ACTIVE
    if any(s == RunStatus.INITIALIZING for s in statuses):
        return RunStatus.INITIALIZING
    return RunStatus.PENDING
