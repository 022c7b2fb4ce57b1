"""Permission engine.

Reference parity: ``vantage6-server/vantage6/server/permission.py`` and
``default_roles.py`` — a rule is a
(resource, scope ∈ {OWN, ORGANIZATION, COLLABORATION, GLOBAL},
operation ∈ {VIEW, EDIT, CREATE, DELETE, SEND, RECEIVE}) triple; roles bundle
rules; default roles (Root, Organization-Admin, Researcher, Viewer) are
seeded at first boot.
"""
from __future__ import annotations

from enum import Enum

from .model import Role, Rule, User


class Scope(str, Enum):
    OWN = "own"
    ORGANIZATION = "organization"
    COLLABORATION = "collaboration"
    GLOBAL = "global"


class Operation(str, Enum):
    VIEW = "view"
    EDIT = "edit"
    CREATE = "create"
    DELETE = "delete"
    SEND = "send"
    RECEIVE = "receive"


RESOURCES = [
    "user",
    "organization",
    "collaboration",
    "node",
    "task",
    "run",
    "result",
    "role",
    "rule",
    "study",
    "event",
    "algorithm",  # algorithm-store entries
]

# scope ordering: a broader scope implies the narrower ones
_SCOPE_ORDER = {Scope.OWN: 0, Scope.ORGANIZATION: 1, Scope.COLLABORATION: 2, Scope.GLOBAL: 3}


def seed_rules(session) -> None:
    """Create the full rule matrix (idempotent)."""
    existing = {(r.name, r.scope, r.operation) for r in session.query(Rule).all()}
    for res in RESOURCES:
        for sc in Scope:
            for op in Operation:
                key = (res, sc.value, op.value)
                if key not in existing:
                    session.add(Rule(name=res, scope=sc.value, operation=op.value))
    session.commit()


DEFAULT_ROLES = {
    "Root": lambda r: True,
    "Organization-Admin": lambda r: r.scope in (Scope.OWN.value, Scope.ORGANIZATION.value),
    "Researcher": lambda r: (
        (r.name in ("task", "run", "result") and r.scope != Scope.GLOBAL.value)
        or (
            r.operation == Operation.VIEW.value
            and r.name in ("organization", "collaboration", "node", "study")
            and r.scope != Scope.GLOBAL.value
        )
        or (r.name == "event" and r.scope != Scope.GLOBAL.value)
    ),
    "Viewer": lambda r: r.operation == Operation.VIEW.value and r.scope != Scope.GLOBAL.value,
}


def seed_default_roles(session) -> None:
    rules = session.query(Rule).all()
    have = {r.name for r in session.query(Role).filter(Role.organization_id.is_(None)).all()}
    for role_name, predicate in DEFAULT_ROLES.items():
        if role_name in have:
            continue
        role = Role(name=role_name, description=f"default role {role_name}")
        role.rules = [r for r in rules if predicate(r)]
        session.add(role)
    session.commit()


class RuleCollection:
    """Per-resource permission checks for one authenticated identity.

    Mirrors the reference's ``RuleCollection`` accessors
    (``self.r.v_glo.can()`` style becomes ``rc.can(Operation.VIEW,
    Scope.GLOBAL)`` / ``rc.allowed_scope(op)``).
    """

    def __init__(self, resource: str, rules: set[tuple[str, str, str]]):
        self.resource = resource
        self._rules = rules

    def can(self, operation: Operation, scope: Scope) -> bool:
        """True if the identity may perform `operation` at >= `scope`."""
        need = _SCOPE_ORDER[scope]
        for res, sc, op in self._rules:
            if res == self.resource and op == operation.value:
                if _SCOPE_ORDER[Scope(sc)] >= need:
                    return True
        return False

    def allowed_scope(self, operation: Operation) -> Scope | None:
        """Widest scope at which `operation` is allowed (None if not at all)."""
        best = None
        for res, sc, op in self._rules:
            if res == self.resource and op == operation.value:
                s = Scope(sc)
                if best is None or _SCOPE_ORDER[s] > _SCOPE_ORDER[best]:
                    best = s
        return best


class PermissionManager:
    """Resolves an identity's effective rule set and hands out collections."""

    def __init__(self, session):
        self.session = session

    def rules_for_user(self, user: User) -> set[tuple[str, str, str]]:
        out: set[tuple[str, str, str]] = set()
        for role in user.roles:
            for rule in role.rules:
                out.add((rule.name, rule.scope, rule.operation))
        return out

    def collection(self, resource: str, user: User) -> RuleCollection:
        return RuleCollection(resource, self.rules_for_user(user))


class PermissionError_(Exception):
    """401/403-class error raised by the service layer."""

    def __init__(self, msg: str, status_code: int = 403):
        super().__init__(msg)
        self.status_code = status_code
