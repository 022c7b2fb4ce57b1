"""Common substrate shared by server, stations, clients and algorithms.

Reference parity: ``vantage6-common/vantage6/common`` —
status enums, serialization, hybrid encryption, configuration management,
logging helpers and base64 utilities.
"""
import base64
import logging


def logger_name(special__name__: str) -> str:
    """Mirror of ``vantage6-common/vantage6/common/__init__.py::logger_name``."""
    module_name = special__name__.split(".")[-1]
    return module_name if module_name != "__main__" else "main"


def bytes_to_base64s(data: bytes) -> str:
    """bytes -> base64 string (``vantage6.common.bytes_to_base64s``)."""
    return base64.b64encode(data).decode("ascii")


def base64s_to_bytes(data: str) -> bytes:
    """base64 string -> bytes (``vantage6.common.base64s_to_bytes``)."""
    return base64.b64decode(data.encode("ascii"))


class Singleton(type):
    """Metaclass making a class a singleton (``vantage6.common.Singleton``)."""

    _instances: dict = {}

    def __call__(cls, *args, **kwargs):
        if cls not in cls._instances:
            cls._instances[cls] = super().__call__(*args, **kwargs)
        return cls._instances[cls]

    @classmethod
    def clear(mcs, target=None):
        """Testing hook: drop cached instance(s)."""
        if target is None:
            mcs._instances.clear()
        else:
            mcs._instances.pop(target, None)


_LOG_FORMAT = "%(asctime)s %(levelname)-8s %(name)s | %(message)s"


def setup_logging(level: int = logging.INFO) -> None:
    logging.basicConfig(level=level, format=_LOG_FORMAT)


def info(msg: str) -> None:
    logging.getLogger("vantage6_amd").info(msg)


def warning(msg: str) -> None:
    logging.getLogger("vantage6_amd").warning(msg)


def error(msg: str) -> None:
    logging.getLogger("vantage6_amd").error(msg)
