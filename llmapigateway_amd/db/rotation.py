"""Round-robin rotation state, persisted in SQLite.

Parity: the reference's ModelRotationDB
(/root/reference/llm_gateway_core/db/model_rotation_db.py:36-110) — state is
keyed by (api_key, gateway_model), advances per request at request start
regardless of outcome, and wraps modulo the number of models. Unlike the
reference (one connection per call), we keep a single connection with a lock,
which is both faster and safe under asyncio worker threads.
"""

from __future__ import annotations

import logging
import os
import sqlite3
import threading
from pathlib import Path

logger = logging.getLogger(__name__)


class ModelRotationDB:
    def __init__(self, db_path: str | os.PathLike = "db/llmgateway_rotation.db"):
        self.db_path = Path(db_path)
        self.db_path.parent.mkdir(parents=True, exist_ok=True)
        self._lock = threading.Lock()
        self._conn = sqlite3.connect(str(self.db_path), check_same_thread=False)
        self._conn.execute(
            """
            CREATE TABLE IF NOT EXISTS model_rotation (
                api_key TEXT NOT NULL,
                gateway_model TEXT NOT NULL,
                last_model_index INTEGER NOT NULL,
                PRIMARY KEY (api_key, gateway_model)
            )
            """
        )
        self._conn.commit()

    def get_next_model_index(self, api_key: str, gateway_model: str, total_models: int) -> int:
        """Return the index to start this request at, then advance the stored state.

        First request for a key/model returns 0; each subsequent request
        returns (previous + 1) % total_models. Any error returns 0 (the
        reference's behavior, model_rotation_db.py:104-110).
        """
        if total_models <= 0:
            return 0
        try:
            with self._lock:
                cur = self._conn.execute(
                    "SELECT last_model_index FROM model_rotation WHERE api_key=? AND gateway_model=?",
                    (api_key, gateway_model),
                )
                row = cur.fetchone()
                next_index = 0 if row is None else (row[0] + 1) % total_models
                self._conn.execute(
                    "INSERT INTO model_rotation (api_key, gateway_model, last_model_index) "
                    "VALUES (?, ?, ?) "
                    "ON CONFLICT(api_key, gateway_model) DO UPDATE SET last_model_index=excluded.last_model_index",
                    (api_key, gateway_model, next_index),
                )
                self._conn.commit()
                return next_index
        except sqlite3.Error:
            logger.exception("rotation DB error; defaulting to index 0")
            return 0

    def close(self) -> None:
        with self._lock:
            self._conn.close()
