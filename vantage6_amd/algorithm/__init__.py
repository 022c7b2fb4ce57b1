"""Algorithm-facing plane.

Reference parity: ``vantage6-algorithm-tools`` —
``AlgorithmClient`` (used inside algorithm code to fan out subtasks and
collect results), the wrapper/dispatch glue (``wrap.py``), the injection
decorators (``@data``, ``@algorithm_client``, ``@metadata``), the
``MockAlgorithmClient`` test vehicle and the typed error hierarchy.

In the MI355X design an "algorithm" is a Python package (not a Docker image)
executed in-process by a GPU-pinned station worker; the I/O contract
(INPUT_FILE/OUTPUT_FILE/TOKEN_FILE + env map) is
preserved so algorithm code is portable between the mock client, the CPU
stations and the GPU stations.
"""
from .exceptions import (  # noqa: F401
    AlgorithmError,
    DataError,
    EnvironmentVariableError,
    InputError,
    UserInputError,
)
from .decorators import algorithm_client, data, metadata  # noqa: F401
from .client import AlgorithmClient  # noqa: F401
from .mock_client import MockAlgorithmClient  # noqa: F401
from .util import error, get_env_var, info, warn  # noqa: F401
