"""Structured JSON logger (parity with the reference's zap wrapper,
logger/logger.go:12-109: dev vs prod config, debug suppressed outside
development, no-op under test)."""
from __future__ import annotations

import json
import sys
import time
from typing import Any


class Logger:
    def __init__(self, environment: str = "production", stream=None,
                 noop: bool = False):
        self.environment = environment
        self.stream = stream or sys.stderr
        self.noop = noop or ("pytest" in sys.modules and stream is None)

    def _emit(self, level: str, msg: str, kv: dict) -> None:
        if self.noop:
            return
        rec = {"ts": round(time.time(), 3), "level": level, "msg": msg}
        rec.update(kv)
        try:
            self.stream.write(json.dumps(rec, default=str) + "\n")
            self.stream.flush()
        except Exception:
            pass

    @staticmethod
    def _kv(args: tuple) -> dict:
        return {str(args[i]): args[i + 1] for i in range(0, len(args) - 1, 2)}

    def debug(self, msg: str, *kv: Any) -> None:
        if self.environment == "development":
            self._emit("debug", msg, self._kv(kv))

    def info(self, msg: str, *kv: Any) -> None:
        self._emit("info", msg, self._kv(kv))

    def warn(self, msg: str, *kv: Any) -> None:
        self._emit("warn", msg, self._kv(kv))

    def error(self, msg: str, *kv: Any) -> None:
        self._emit("error", msg, self._kv(kv))


def noop_logger() -> Logger:
    return Logger(noop=True)
