"""Structured logging (reference: observability/logging.rs — JSON logs with
request correlation, --log-dir file output, middleware logging layer)."""
from __future__ import annotations

import json
import logging
import os
import sys
import time
from typing import Optional


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out = {
            "ts": round(time.time(), 4),
            "level": record.levelname.lower(),
            "logger": record.name,
            "msg": record.getMessage(),
        }
        for key in ("request_id", "tenant_id", "worker", "model"):
            v = getattr(record, key, None)
            if v is not None:
                out[key] = v
        if record.exc_info:
            out["exc"] = self.formatException(record.exc_info)
        return json.dumps(out)


def setup_logging(level: str = "info", log_json: bool = False, log_dir: Optional[str] = None) -> None:
    root = logging.getLogger()
    root.setLevel(getattr(logging, level.upper(), logging.INFO))
    for h in list(root.handlers):
        root.removeHandler(h)
    handlers = [logging.StreamHandler(sys.stderr)]
    if log_dir:
        os.makedirs(log_dir, exist_ok=True)
        handlers.append(logging.FileHandler(os.path.join(log_dir, "smg.log")))
    fmt = (
        JsonFormatter()
        if log_json
        else logging.Formatter("%(asctime)s %(levelname)s %(name)s %(message)s")
    )
    for h in handlers:
        h.setFormatter(fmt)
        root.addHandler(h)


class RequestLogAdapter(logging.LoggerAdapter):
    """Correlates log lines with a request id: log = RequestLogAdapter(logger, rid)."""

    def __init__(self, logger, request_id: str, tenant_id: Optional[str] = None):
        super().__init__(logger, {"request_id": request_id, "tenant_id": tenant_id})

    def process(self, msg, kwargs):
        extra = kwargs.setdefault("extra", {})
        extra.update(self.extra)
        return msg, kwargs
