"""Persistence & domain model.

Reference parity: ``vantage6-server/vantage6/server/model/`` (SURVEY.md §3.2,
L1) — one ORM class per entity with the reference's field inventory:
``Authenticatable`` base (User + Node share type/ip/last_seen),
``Organization`` (public_key), ``Collaboration`` (encrypted flag, m2m orgs),
``User``, ``Role``/``Rule`` permission matrix, ``Node`` (api_key,
org+collaboration), ``Task`` (image, parent, job_id, databases, init org/user,
study), ``Run`` (input/result b64, log, status, timestamps), ``Study``.

SQLite (file or in-memory) via SQLAlchemy; tasks/runs survive orchestrator
restarts like the reference's server DB does.
"""
from __future__ import annotations

import datetime as dt
import hashlib
import hmac
import secrets

from sqlalchemy import (
    Boolean,
    Column,
    DateTime,
    ForeignKey,
    Integer,
    String,
    Table,
    Text,
    create_engine,
)
from sqlalchemy.orm import declarative_base, relationship, sessionmaker, scoped_session

Base = declarative_base()


def utcnow() -> dt.datetime:
    return dt.datetime.now(dt.timezone.utc).replace(tzinfo=None)


# -- association tables -----------------------------------------------------
Member = Table(
    "member",
    Base.metadata,
    Column("organization_id", ForeignKey("organization.id"), primary_key=True),
    Column("collaboration_id", ForeignKey("collaboration.id"), primary_key=True),
)

role_rule = Table(
    "role_rule",
    Base.metadata,
    Column("role_id", ForeignKey("role.id"), primary_key=True),
    Column("rule_id", ForeignKey("rule.id"), primary_key=True),
)

user_role = Table(
    "user_role",
    Base.metadata,
    Column("user_id", ForeignKey("user.id"), primary_key=True),
    Column("role_id", ForeignKey("role.id"), primary_key=True),
)

study_organization = Table(
    "study_organization",
    Base.metadata,
    Column("study_id", ForeignKey("study.id"), primary_key=True),
    Column("organization_id", ForeignKey("organization.id"), primary_key=True),
)


class Organization(Base):
    __tablename__ = "organization"
    id = Column(Integer, primary_key=True)
    name = Column(String, unique=True, nullable=False)
    domain = Column(String, default="")
    address1 = Column(String, default="")
    country = Column(String, default="")
    public_key = Column(Text, default="")  # base64 public key for E2E encryption

    collaborations = relationship("Collaboration", secondary=Member, back_populates="organizations")
    nodes = relationship("Node", back_populates="organization")
    users = relationship("User", back_populates="organization")


class Collaboration(Base):
    __tablename__ = "collaboration"
    id = Column(Integer, primary_key=True)
    name = Column(String, unique=True, nullable=False)
    encrypted = Column(Boolean, default=False)

    organizations = relationship("Organization", secondary=Member, back_populates="collaborations")
    nodes = relationship("Node", back_populates="collaboration")
    tasks = relationship("Task", back_populates="collaboration")
    studies = relationship("Study", back_populates="collaboration")


class Study(Base):
    """Subset of a collaboration's organizations (vantage6 4.3+)."""

    __tablename__ = "study"
    id = Column(Integer, primary_key=True)
    name = Column(String, nullable=False)
    collaboration_id = Column(Integer, ForeignKey("collaboration.id"))

    collaboration = relationship("Collaboration", back_populates="studies")
    organizations = relationship("Organization", secondary=study_organization)


def hash_secret(secret: str, salt: str) -> str:
    return hashlib.pbkdf2_hmac("sha256", secret.encode(), salt.encode(), 50_000).hex()


class Authenticatable(Base):
    """Base for entities that can hold a JWT (User and Node)."""

    __tablename__ = "authenticatable"
    id = Column(Integer, primary_key=True)
    type = Column(String(50))
    ip = Column(String, default="")
    last_seen = Column(DateTime, nullable=True)
    status = Column(String, default="offline")  # node online/offline tracking

    __mapper_args__ = {"polymorphic_identity": "authenticatable", "polymorphic_on": type}


class User(Authenticatable):
    __tablename__ = "user"
    id = Column(Integer, ForeignKey("authenticatable.id"), primary_key=True)
    username = Column(String, unique=True, nullable=False)
    password_hash = Column(String, default="")
    salt = Column(String, default="")
    firstname = Column(String, default="")
    lastname = Column(String, default="")
    email = Column(String, default="")
    failed_login_attempts = Column(Integer, default=0)
    last_login_attempt = Column(DateTime, nullable=True)
    mfa_secret = Column(String, default="")  # TOTP secret (empty = MFA off)
    organization_id = Column(Integer, ForeignKey("organization.id"))

    organization = relationship("Organization", back_populates="users")
    roles = relationship("Role", secondary=user_role, back_populates="users")

    __mapper_args__ = {"polymorphic_identity": "user"}

    def set_password(self, password: str) -> None:
        self.salt = secrets.token_hex(8)
        self.password_hash = hash_secret(password, self.salt)

    def check_password(self, password: str) -> bool:
        return hmac.compare_digest(self.password_hash, hash_secret(password, self.salt or ""))


class Node(Authenticatable):
    """A data station. In the MI355X deployment one Node row = one GPU-pinned
    in-process station worker (reference: one vantage6 node daemon per org)."""

    __tablename__ = "node"
    id = Column(Integer, ForeignKey("authenticatable.id"), primary_key=True)
    name = Column(String, unique=True, nullable=False)
    api_key_hash = Column(String, default="")
    api_key_salt = Column(String, default="")
    collaboration_id = Column(Integer, ForeignKey("collaboration.id"))
    organization_id = Column(Integer, ForeignKey("organization.id"))
    device = Column(String, default="cpu")  # "cpu" or "cuda:<i>" (HIP device)
    # node-info-update payload (reference: shared node config/capabilities;
    # here: database labels -> column names, device info)
    info_json = Column(Text, default="")

    collaboration = relationship("Collaboration", back_populates="nodes")
    organization = relationship("Organization", back_populates="nodes")

    __mapper_args__ = {"polymorphic_identity": "node"}

    def set_api_key(self, api_key: str) -> None:
        self.api_key_salt = secrets.token_hex(8)
        self.api_key_hash = hash_secret(api_key, self.api_key_salt)

    def check_api_key(self, api_key: str) -> bool:
        return hmac.compare_digest(
            self.api_key_hash, hash_secret(api_key, self.api_key_salt or "")
        )


class Role(Base):
    __tablename__ = "role"
    id = Column(Integer, primary_key=True)
    name = Column(String, nullable=False)
    description = Column(String, default="")
    organization_id = Column(Integer, ForeignKey("organization.id"), nullable=True)

    rules = relationship("Rule", secondary=role_rule, back_populates="roles")
    users = relationship("User", secondary=user_role, back_populates="roles")


class Rule(Base):
    """(resource, scope, operation) permission triple."""

    __tablename__ = "rule"
    id = Column(Integer, primary_key=True)
    name = Column(String, nullable=False)  # resource name, e.g. "task"
    scope = Column(String, nullable=False)  # own|organization|collaboration|global
    operation = Column(String, nullable=False)  # view|edit|create|delete|send|receive

    roles = relationship("Role", secondary=role_rule, back_populates="rules")


class Task(Base):
    __tablename__ = "task"
    id = Column(Integer, primary_key=True)
    name = Column(String, default="")
    description = Column(String, default="")
    image = Column(String, nullable=False)  # algorithm package ref (reference: docker image)
    method = Column(String, default="")
    collaboration_id = Column(Integer, ForeignKey("collaboration.id"))
    study_id = Column(Integer, ForeignKey("study.id"), nullable=True)
    parent_id = Column(Integer, ForeignKey("task.id"), nullable=True)
    job_id = Column(Integer, default=0)  # shared by a task tree (central + partials)
    init_org_id = Column(Integer, ForeignKey("organization.id"), nullable=True)
    init_user_id = Column(Integer, ForeignKey("user.id"), nullable=True)
    databases = Column(Text, default="[]")  # JSON list of {"label": ...}
    created_at = Column(DateTime, default=utcnow)

    collaboration = relationship("Collaboration", back_populates="tasks")
    parent = relationship("Task", remote_side=[id], backref="children")
    runs = relationship("Run", back_populates="task", cascade="all, delete-orphan")


class Run(Base):
    """One organization's execution of a task (v3 name: Result)."""

    __tablename__ = "run"
    id = Column(Integer, primary_key=True)
    task_id = Column(Integer, ForeignKey("task.id"))
    organization_id = Column(Integer, ForeignKey("organization.id"))
    node_id = Column(Integer, ForeignKey("node.id"), nullable=True)
    status = Column(String, default="pending")
    input = Column(Text, default="")  # b64 (possibly encrypted) input
    result = Column(Text, default="")  # b64 (possibly encrypted) result
    log = Column(Text, default="")
    assigned_at = Column(DateTime, default=utcnow)
    started_at = Column(DateTime, nullable=True)
    finished_at = Column(DateTime, nullable=True)
    ports = Column(Text, default="[]")  # JSON list (reference: VPN Port rows)

    task = relationship("Task", back_populates="runs")
    organization = relationship("Organization")


class Algorithm(Base):
    """Algorithm-store entry (vantage6-algorithm-store, 4.2+):
    curated registry of algorithm packages with a review workflow
    (submitted -> approved/rejected); stations can require approval."""

    __tablename__ = "algorithm"
    id = Column(Integer, primary_key=True)
    name = Column(String, nullable=False)
    image = Column(String, unique=True, nullable=False)  # module path
    description = Column(String, default="")
    functions = Column(Text, default="[]")  # JSON: [{name, args, doc}]
    status = Column(String, default="submitted")  # submitted|approved|rejected
    submitted_by = Column(Integer, ForeignKey("user.id"), nullable=True)
    reviewed_by = Column(Integer, ForeignKey("user.id"), nullable=True)
    review_comment = Column(String, default="")
    created_at = Column(DateTime, default=utcnow)


class Database:
    """Engine/session holder (``vantage6-server/.../model/base.py::Database``)."""

    def __init__(self, uri: str = "sqlite://"):
        kwargs = {"future": True, "connect_args": {"check_same_thread": False}}
        if uri in ("sqlite://", "sqlite:///:memory:", ":memory:"):
            # in-memory SQLite: every connection is a fresh empty DB, so all
            # threads must share ONE connection (sqlite3 serializes it)
            from sqlalchemy.pool import StaticPool

            uri = "sqlite://"
            kwargs["poolclass"] = StaticPool
        elif not uri.startswith("sqlite"):
            uri = f"sqlite:///{uri}"
        self.engine = create_engine(uri, **kwargs)
        Base.metadata.create_all(self.engine)
        self.session_factory = scoped_session(
            sessionmaker(bind=self.engine, expire_on_commit=False, future=True)
        )

    def session(self):
        return self.session_factory()

    def close(self) -> None:
        self.session_factory.remove()
        self.engine.dispose()
