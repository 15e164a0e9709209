"""Structured JSONL metrics + the reference's human-readable epoch line."""
from __future__ import annotations

import json
import os
import time


class JsonlLogger:
    def __init__(self, path: str | None, rank: int = 0):
        self.rank = rank
        self.f = None
        if path and rank == 0:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self.f = open(path, "a", buffering=1)

    def log(self, record: dict):
        if self.f is not None:
            record = {"ts": time.time(), **record}
            self.f.write(json.dumps(record) + "\n")

    def print0(self, msg: str):
        if self.rank == 0:
            print(msg, flush=True)

    def close(self):
        if self.f is not None:
            self.f.close()
            self.f = None
