"""Colored, per-module-level logging (capability parity with the reference's
parallax_utils/logging_config.py; re-designed, not a copy)."""

import logging
import os
import sys

_RESET = "\x1b[0m"
_COLORS = {
    logging.DEBUG: "\x1b[36m",
    logging.INFO: "\x1b[32m",
    logging.WARNING: "\x1b[33m",
    logging.ERROR: "\x1b[31m",
    logging.CRITICAL: "\x1b[41m",
}


class _ColorFormatter(logging.Formatter):
    def format(self, record):
        base = super().format(record)
        if sys.stderr.isatty():
            color = _COLORS.get(record.levelno, "")
            return f"{color}{base}{_RESET}"
        return base


_configured = False


def get_logger(name: str) -> logging.Logger:
    global _configured
    if not _configured:
        handler = logging.StreamHandler(sys.stderr)
        handler.setFormatter(
            _ColorFormatter(
                fmt="%(asctime)s [%(levelname)s] %(name)s: %(message)s",
                datefmt="%H:%M:%S",
            )
        )
        root = logging.getLogger("parallax_amd")
        root.addHandler(handler)
        root.setLevel(os.environ.get("PARALLAX_AMD_LOG_LEVEL", "INFO").upper())
        root.propagate = False
        _configured = True
    return logging.getLogger(name if name.startswith("parallax_amd") else f"parallax_amd.{name}")


def set_module_level(module: str, level: str) -> None:
    logging.getLogger(f"parallax_amd.{module}").setLevel(level.upper())
