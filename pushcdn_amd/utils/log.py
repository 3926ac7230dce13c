"""Structured logging (reference L8: `tracing` with env-filter + optional
JSON output; binaries/broker.rs:80-93).

Env vars mirror the reference's: ``CDN_LOG`` (level filter, default INFO) and
``CDN_LOG_FORMAT=json`` for JSON lines.  Identities are logged as mnemonics
(utils/mnemonic.py), like the reference's hash-derived names (util.rs:13-23).
"""

from __future__ import annotations

import json
import logging
import os
import sys
import time

from .mnemonic import mnemonic


class _JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out = {
            "ts": round(time.time(), 3),
            "level": record.levelname,
            "target": record.name,
            "msg": record.getMessage(),
        }
        extra = getattr(record, "fields", None)
        if extra:
            out.update(extra)
        return json.dumps(out)


_configured = False


def _configure() -> None:
    global _configured
    if _configured:
        return
    _configured = True
    level = os.environ.get("CDN_LOG", "INFO").upper()
    handler = logging.StreamHandler(sys.stderr)
    if os.environ.get("CDN_LOG_FORMAT") == "json":
        handler.setFormatter(_JsonFormatter())
    else:
        handler.setFormatter(
            logging.Formatter("%(asctime)s %(levelname)-5s %(name)s: %(message)s")
        )
    root = logging.getLogger("pushcdn")
    root.addHandler(handler)
    try:
        root.setLevel(level)
    except ValueError:
        root.setLevel(logging.INFO)


def get_logger(name: str) -> logging.Logger:
    _configure()
    return logging.getLogger(f"pushcdn.{name}")


def ident(identity: bytes) -> str:
    """Human-readable identity mnemonic for log fields."""
    return mnemonic(identity)
