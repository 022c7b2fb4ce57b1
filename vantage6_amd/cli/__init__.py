"""Ops / CLI plane (``v6`` command analog) — see ``vantage6_amd.cli.main``."""
