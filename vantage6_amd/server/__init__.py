"""Orchestrator ("server") plane.

Reference parity: ``vantage6-server`` — the Flask REST API +
SQLAlchemy domain model + Socket.IO eventing + permission engine, collapsed
into an in-process orchestrator for a single 8×MI355X node:

- :mod:`vantage6_amd.server.model` — the SQLAlchemy entities (Task, Run,
  Node, Organization, Collaboration, User, Role, Rule, Study).
- :mod:`vantage6_amd.server.permission` — rule/scope/operation engine.
- :mod:`vantage6_amd.server.core` — the service layer (task brokering, auth,
  eventing); both the in-process client and the FastAPI façade call this.
- :mod:`vantage6_amd.server.api` — FastAPI façade speaking the vantage6
  client JSON schema.
"""
from .core import ServerCore  # noqa: F401
