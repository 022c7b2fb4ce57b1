"""Station ("node") plane.

Reference parity: ``vantage6-node`` — the node daemon that
authenticates with an api key, listens for tasks, runs algorithms and
reports results. The Docker/VPN/squid machinery of the reference is replaced
by in-process execution on a pinned HIP device: one :class:`Station` per
GPU (or CPU for tests), algorithms are Python packages, and the node-local
proxy server becomes :class:`vantage6_amd.node.proxy.NodeProxy`.
"""
from .station import Station  # noqa: F401
from .proxy import NodeProxy  # noqa: F401
