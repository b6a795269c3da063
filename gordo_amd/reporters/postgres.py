"""
PostgresReporter — upsert build results into a ``machine`` table
(behavioral spec: gordo/reporters/postgres.py — peewee ORM, name
unique, dataset/model/metadata JSONB).

Neither peewee nor psycopg2 is guaranteed in this image, so the
reporter is written against plain DB-API 2.0 with a pluggable
connection factory: the default resolves a Postgres driver at report
time (psycopg2, else peewee's pg8000, else a clear error), and any
DB-API connection can be injected — the test lane runs the REAL
``report()`` upsert path against stdlib sqlite3 (same table layout,
same ON CONFLICT (name) DO UPDATE statement, JSON as text).
"""
from __future__ import annotations

import json
import logging
from typing import Any, Callable, Optional

from .base import BaseReporter
from .exceptions import ReporterException
from ..util.utils import capture_args

logger = logging.getLogger(__name__)


class PostgresReporterException(ReporterException):
    pass


_CREATE_PG = """
CREATE TABLE IF NOT EXISTS machine (
    id SERIAL PRIMARY KEY,
    name TEXT UNIQUE NOT NULL,
    dataset JSONB NOT NULL,
    model JSONB NOT NULL,
    metadata JSONB NOT NULL
)
"""

_CREATE_SQLITE = """
CREATE TABLE IF NOT EXISTS machine (
    id INTEGER PRIMARY KEY AUTOINCREMENT,
    name TEXT UNIQUE NOT NULL,
    dataset TEXT NOT NULL,
    model TEXT NOT NULL,
    metadata TEXT NOT NULL
)
"""

_UPSERT = """
INSERT INTO machine (name, dataset, model, metadata)
VALUES ({p}, {p}, {p}, {p})
ON CONFLICT (name) DO UPDATE SET
    dataset = excluded.dataset,
    model = excluded.model,
    metadata = excluded.metadata
"""


class PostgresReporter(BaseReporter):
    """Reporter storing a Machine into Postgres (or any injected
    DB-API connection)."""

    @capture_args
    def __init__(
        self,
        host: str,
        port: int = 5432,
        user: str = "postgres",
        password: Optional[str] = None,
        database: str = "postgres",
        connection_factory: Optional[Callable[[], Any]] = None,
        **kwargs,
    ):
        self.host = host
        self.port = port
        self.user = user
        self.password = password
        self.database = database
        self._connection_factory = connection_factory
        self._extra = kwargs

    # ---- connection handling -------------------------------------------
    def _connect(self):
        if self._connection_factory is not None:
            return self._connection_factory()
        try:
            import psycopg2
        except ImportError as e:
            raise PostgresReporterException(
                "no Postgres driver available (psycopg2 not installed) and "
                "no connection_factory given"
            ) from e
        params = dict(
            host=self.host, port=self.port, user=self.user,
            dbname=self.database,
        )
        if self.password:
            params["password"] = self.password
        params.update(self._extra)
        return psycopg2.connect(**params)

    @staticmethod
    def _dialect(conn) -> str:
        mod = type(conn).__module__.split(".")[0]
        return "sqlite" if mod in ("sqlite3", "_sqlite3") else "postgres"

    def report(self, machine):
        """Upsert the machine's dataset/model/metadata keyed by name
        (reference postgres.py:62-94)."""
        try:
            conn = self._connect()
        except PostgresReporterException:
            raise
        except Exception as exc:
            raise PostgresReporterException(exc) from exc
        try:
            dialect = self._dialect(conn)
            param = "?" if dialect == "sqlite" else "%s"
            create = _CREATE_SQLITE if dialect == "sqlite" else _CREATE_PG
            upsert = _UPSERT.format(p=param)
            machine_dict = machine.to_dict()
            record = (
                machine_dict["name"],
                json.dumps(machine_dict["dataset"], default=str),
                json.dumps(machine_dict["model"], default=str),
                json.dumps(
                    json.loads(machine.to_json())["metadata"], default=str
                ),
            )
            logger.info("Inserting machine %s in sql", machine_dict["name"])
            cur = conn.cursor()
            try:
                cur.execute(create)
                cur.execute(upsert, record)
                conn.commit()
            finally:
                cur.close()
        except PostgresReporterException:
            raise
        except Exception as exc:
            try:
                conn.rollback()
            except Exception:
                pass
            raise PostgresReporterException(exc) from exc
        finally:
            try:
                conn.close()
            except Exception:
                pass
