"""Declarative data preprocessing before the algorithm sees the data.

Reference parity: ``vantage6-algorithm-tools/.../preprocessing.py``
 — a task's database spec may carry a ``preprocessing``
list; each step is applied to the loaded DataFrame before injection:

    {"label": "default", "preprocessing": [
        {"function": "select_columns", "parameters": {"columns": ["age"]}},
        {"function": "filter_range", "parameters": {"column": "age", "min": 18}},
    ]}
"""
from __future__ import annotations

from typing import Any

from .exceptions import UserInputError


def select_columns(df, columns: list[str]):
    missing = [c for c in columns if c not in df.columns]
    if missing:
        raise UserInputError(f"columns not in data: {missing}")
    return df[columns]


def drop_columns(df, columns: list[str]):
    return df.drop(columns=[c for c in columns if c in df.columns])


def filter_range(df, column: str, min: Any = None, max: Any = None):  # noqa: A002
    if column not in df.columns:
        raise UserInputError(f"column {column!r} not in data")
    out = df
    if min is not None:
        out = out[out[column] >= min]
    if max is not None:
        out = out[out[column] <= max]
    return out


def filter_rows(df, column: str, values: list):
    if column not in df.columns:
        raise UserInputError(f"column {column!r} not in data")
    return df[df[column].isin(values)]


def dropna(df, columns: list[str] | None = None):
    return df.dropna(subset=columns)


_FUNCTIONS = {
    "select_columns": select_columns,
    "drop_columns": drop_columns,
    "filter_range": filter_range,
    "filter_rows": filter_rows,
    "dropna": dropna,
}


def apply_preprocessing(df, steps: list[dict]):
    """Apply an ordered list of {"function", "parameters"} steps."""
    for step in steps or []:
        name = step.get("function")
        fn = _FUNCTIONS.get(name)
        if fn is None:
            raise UserInputError(
                f"unknown preprocessing function {name!r} "
                f"(available: {sorted(_FUNCTIONS)})"
            )
        df = fn(df, **(step.get("parameters") or {}))
    return df
