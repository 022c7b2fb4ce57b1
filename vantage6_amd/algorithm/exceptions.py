"""Typed algorithm error hierarchy.

Reference parity: ``vantage6-algorithm-tools/vantage6/algorithm/tools/
exceptions.py`` — errors raised inside algorithm code map to
run failure statuses on the server.
"""


class AlgorithmError(Exception):
    """Base error raised by algorithm code; run status becomes 'failed'."""


class InputError(AlgorithmError):
    """Malformed INPUT_FILE contents / method signature mismatch."""


class UserInputError(InputError):
    """The researcher supplied invalid arguments."""


class EnvironmentVariableError(AlgorithmError):
    """Required env entry missing from the runtime contract."""


class DataError(AlgorithmError):
    """A station database could not be loaded."""


class CollectResultsError(AlgorithmError):
    """Subtask result collection failed (a partial crashed or was killed)."""


class PrivacyThresholdViolation(AlgorithmError):
    """A station refused to answer because its shard is below the privacy
    threshold (reference: disclosure checks in community algorithms such as
    v6-summary-statistics)."""
