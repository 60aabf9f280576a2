"""Shared-memory broadcast — the ``sc.broadcast`` replacement.

The reference ships medium-size datasets to workers via Spark's torrent
broadcast (``hyperopt/2. hyperopt on diff sizes of data.py:92-99``:
``bc = sc.broadcast(data); bc.value``). On one node the workers are local
processes, so the broadcast is a file in /dev/shm: pickled once by the
producer, lazily unpickled (page-cache shared) by each consumer. The
``.value`` / ``.unpersist()`` API is preserved.
"""
from __future__ import annotations

import os
import pickle
import tempfile
import uuid
from typing import Any, Optional

_SHM_DIR = "/dev/shm" if os.path.isdir("/dev/shm") else tempfile.gettempdir()


class Broadcast:
    def __init__(self, path: str):
        self.path = path
        self._value: Optional[Any] = None
        self._loaded = False

    @property
    def value(self) -> Any:
        if not self._loaded:
            with open(self.path, "rb") as f:
                self._value = pickle.load(f)
            self._loaded = True
        return self._value

    def unpersist(self) -> None:
        self._value = None
        self._loaded = False
        try:
            os.unlink(self.path)
        except FileNotFoundError:
            pass

    # pickling a Broadcast ships only the shm path, not the payload —
    # exactly the property Spark broadcast gives closures.
    def __getstate__(self):
        return {"path": self.path}

    def __setstate__(self, state):
        self.path = state["path"]
        self._value = None
        self._loaded = False


def broadcast(obj: Any) -> Broadcast:
    path = os.path.join(_SHM_DIR, f"mi355x-bcast-{uuid.uuid4().hex}.pkl")
    with open(path, "wb") as f:
        pickle.dump(obj, f, protocol=pickle.HIGHEST_PROTOCOL)
    return Broadcast(path)
