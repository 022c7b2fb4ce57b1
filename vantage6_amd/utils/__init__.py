"""Shared utilities (metrics/observability)."""
