"""File + ANSI-colour console logger (same API surface as reference src/Log.py:15-50)."""

from __future__ import annotations

import logging
import os
import sys

_COLORS = {
    "red": "\033[91m",
    "green": "\033[92m",
    "yellow": "\033[93m",
    "blue": "\033[94m",
    "magenta": "\033[95m",
    "cyan": "\033[96m",
    "white": "\033[97m",
}
_RESET = "\033[0m"


def print_with_color(text: str, color: str = "white") -> None:
    code = _COLORS.get(color, _COLORS["white"])
    stream = sys.stdout
    if stream.isatty():
        print(f"{code}{text}{_RESET}")
    else:
        print(text)


class Logger:
    """Writes to a log file always; echoes debug messages when debug_mode is on."""

    def __init__(self, log_path: str, debug_mode: bool = False):
        self.debug_mode = debug_mode
        os.makedirs(os.path.dirname(log_path) or ".", exist_ok=True)
        self._log = logging.getLogger(f"split_learning_amd.{log_path}")
        self._log.setLevel(logging.DEBUG)
        if not self._log.handlers:
            fh = logging.FileHandler(log_path)
            fh.setFormatter(logging.Formatter("%(asctime)s [%(levelname)s] %(message)s"))
            self._log.addHandler(fh)
        self._log.propagate = False

    def log_info(self, message: str) -> None:
        self._log.info(message)
        if self.debug_mode:
            print_with_color(f"[INFO] {message}", "green")

    def log_warning(self, message: str) -> None:
        self._log.warning(message)
        if self.debug_mode:
            print_with_color(f"[WARN] {message}", "yellow")

    def log_error(self, message: str) -> None:
        self._log.error(message)
        print_with_color(f"[ERROR] {message}", "red")

    def log_debug(self, message: str) -> None:
        self._log.debug(message)
        if self.debug_mode:
            print_with_color(f"[DEBUG] {message}", "cyan")
