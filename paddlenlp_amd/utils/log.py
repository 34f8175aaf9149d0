"""Singleton colored logger (reference: paddlenlp/utils/log.py)."""
import logging
import os
import sys
import threading

_LOG_LEVEL = os.environ.get("PDNLP_AMD_LOG_LEVEL", "INFO").upper()

_COLORS = {
    "DEBUG": "\033[37m",
    "INFO": "\033[32m",
    "WARNING": "\033[33m",
    "ERROR": "\033[31m",
    "CRITICAL": "\033[35m",
}
_RESET = "\033[0m"


class _ColorFormatter(logging.Formatter):
    def format(self, record):
        msg = super().format(record)
        if sys.stderr.isatty():
            color = _COLORS.get(record.levelname, "")
            return f"{color}{msg}{_RESET}"
        return msg


class Logger:
    _instance = None
    _lock = threading.Lock()

    def __new__(cls):
        with cls._lock:
            if cls._instance is None:
                cls._instance = super().__new__(cls)
                cls._instance._init()
        return cls._instance

    def _init(self):
        self._logger = logging.getLogger("paddlenlp_amd")
        self._logger.propagate = False
        if not self._logger.handlers:
            handler = logging.StreamHandler(sys.stderr)
            handler.setFormatter(
                _ColorFormatter("[%(asctime)s] [%(levelname)8s] - %(message)s", "%Y-%m-%d %H:%M:%S")
            )
            self._logger.addHandler(handler)
        self._logger.setLevel(_LOG_LEVEL)

    def set_level(self, level):
        self._logger.setLevel(level)

    def __getattr__(self, name):
        return getattr(self._logger, name)


logger = Logger()
