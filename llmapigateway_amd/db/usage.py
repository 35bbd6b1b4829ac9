"""Token-usage accounting, persisted in SQLite.

Parity: the reference's TokensUsageDB
(/root/reference/llm_gateway_core/db/tokens_usage_db.py) — same table shape
(prompt/completion/total/reasoning/cached tokens, cost, model, provider with
a timestamp index), same aggregation buckets (hour/day/week/month via
strftime, tokens_usage_db.py:242-252), paginated latest-records fetch, total
count, and a 180-day cleanup. Single locked connection instead of
one-connection-per-call.
"""

from __future__ import annotations

import logging
import os
import sqlite3
import threading
from datetime import datetime, timedelta
from pathlib import Path
from typing import Any, Dict, List, Optional

logger = logging.getLogger(__name__)

_PERIOD_FORMATS = {
    "hour": "%Y-%m-%d %H:00:00",
    "day": "%Y-%m-%d",
    "week": "%Y-W%W",
    "month": "%Y-%m",
}


class TokensUsageDB:
    def __init__(self, db_path: str | os.PathLike = "db/tokens_usage.db"):
        self.db_path = Path(db_path)
        self.db_path.parent.mkdir(parents=True, exist_ok=True)
        self._lock = threading.Lock()
        self._conn = sqlite3.connect(str(self.db_path), check_same_thread=False)
        self._conn.execute(
            """
            CREATE TABLE IF NOT EXISTS tokens_usage (
                id INTEGER PRIMARY KEY AUTOINCREMENT,
                timestamp TEXT NOT NULL,
                prompt_tokens INTEGER NOT NULL DEFAULT 0,
                completion_tokens INTEGER NOT NULL DEFAULT 0,
                total_tokens INTEGER NOT NULL DEFAULT 0,
                reasoning_tokens INTEGER NOT NULL DEFAULT 0,
                cached_tokens INTEGER NOT NULL DEFAULT 0,
                cost REAL NOT NULL DEFAULT 0.0,
                model TEXT,
                provider TEXT
            )
            """
        )
        self._conn.execute(
            "CREATE INDEX IF NOT EXISTS idx_tokens_usage_timestamp ON tokens_usage (timestamp)"
        )
        self._conn.commit()

    def insert_usage(
        self,
        prompt_tokens: int = 0,
        completion_tokens: int = 0,
        total_tokens: Optional[int] = None,
        reasoning_tokens: int = 0,
        cached_tokens: int = 0,
        cost: float = 0.0,
        model: Optional[str] = None,
        provider: Optional[str] = None,
        timestamp: Optional[datetime] = None,
    ) -> bool:
        ts = (timestamp or datetime.now()).isoformat()
        if total_tokens is None:
            total_tokens = prompt_tokens + completion_tokens
        try:
            with self._lock:
                self._conn.execute(
                    "INSERT INTO tokens_usage (timestamp, prompt_tokens, completion_tokens, "
                    "total_tokens, reasoning_tokens, cached_tokens, cost, model, provider) "
                    "VALUES (?,?,?,?,?,?,?,?,?)",
                    (
                        ts,
                        int(prompt_tokens or 0),
                        int(completion_tokens or 0),
                        int(total_tokens or 0),
                        int(reasoning_tokens or 0),
                        int(cached_tokens or 0),
                        float(cost or 0.0),
                        model,
                        provider,
                    ),
                )
                self._conn.commit()
            return True
        except sqlite3.Error:
            logger.exception("failed to insert usage record")
            return False

    def get_aggregated_usage(
        self,
        period: str,
        start_date: Optional[datetime] = None,
        end_date: Optional[datetime] = None,
    ) -> List[Dict[str, Any]]:
        fmt = _PERIOD_FORMATS.get(period)
        if fmt is None:
            logger.error("Invalid aggregation period: %s", period)
            return []
        where, params = "", []
        if start_date is not None:
            where = " WHERE timestamp >= ?"
            params.append(start_date.isoformat())
        if end_date is not None:
            where += " AND" if where else " WHERE"
            where += " timestamp <= ?"
            params.append(end_date.isoformat())
        query = (
            f"SELECT strftime('{fmt}', timestamp) AS time_period, model, "
            "SUM(prompt_tokens) AS prompt_tokens, SUM(completion_tokens) AS completion_tokens, "
            "SUM(total_tokens) AS total_tokens, SUM(reasoning_tokens) AS reasoning_tokens, "
            "SUM(cached_tokens) AS cached_tokens, SUM(cost) AS cost, COUNT(*) AS count "
            f"FROM tokens_usage{where} GROUP BY time_period, model "
            "ORDER BY time_period DESC, model ASC"
        )
        try:
            with self._lock:
                cur = self._conn.execute(query, params)
                cols = [d[0] for d in cur.description]
                return [dict(zip(cols, row)) for row in cur.fetchall()]
        except sqlite3.Error:
            logger.exception("failed to aggregate usage")
            return []

    def get_latest_usage_records(self, limit: int = 25, offset: int = 0) -> List[Dict[str, Any]]:
        try:
            with self._lock:
                cur = self._conn.execute(
                    "SELECT * FROM tokens_usage ORDER BY timestamp DESC, id DESC LIMIT ? OFFSET ?",
                    (int(limit), int(offset)),
                )
                cols = [d[0] for d in cur.description]
                return [dict(zip(cols, row)) for row in cur.fetchall()]
        except sqlite3.Error:
            logger.exception("failed to fetch usage records")
            return []

    def get_total_records_count(self) -> int:
        try:
            with self._lock:
                cur = self._conn.execute("SELECT COUNT(*) FROM tokens_usage")
                return int(cur.fetchone()[0])
        except sqlite3.Error:
            logger.exception("failed to count usage records")
            return 0

    def cleanup_old_records(self, days: int = 180) -> int:
        """Delete records older than `days`. Unlike the reference (where this
        exists but is never called, tokens_usage_db.py:164), the app invokes
        it at startup."""
        cutoff = (datetime.now() - timedelta(days=days)).isoformat()
        try:
            with self._lock:
                cur = self._conn.execute("DELETE FROM tokens_usage WHERE timestamp < ?", (cutoff,))
                self._conn.commit()
                return cur.rowcount
        except sqlite3.Error:
            logger.exception("failed to clean up usage records")
            return 0

    def close(self) -> None:
        with self._lock:
            self._conn.close()
