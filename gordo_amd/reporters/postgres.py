"""
PostgresReporter — upsert build results into a ``machine`` table
(behavioral spec: gordo/reporters/postgres.py — peewee ORM, name
unique, dataset/model/metadata JSONB).

peewee is not installed in every environment; when missing, a direct
psycopg2/sqlite-free fallback is unavailable and constructing the
reporter raises a clear error at report time (configs still parse).
"""
from __future__ import annotations

import json
import logging
from typing import Optional

from .base import BaseReporter
from .exceptions import ReporterException
from ..util.utils import capture_args

logger = logging.getLogger(__name__)

try:
    import peewee
    from playhouse.postgres_ext import BinaryJSONField, PostgresqlExtDatabase

    HAS_PEEWEE = True
except ImportError:
    HAS_PEEWEE = False


class PostgresReporterException(ReporterException):
    pass


if HAS_PEEWEE:
    db = PostgresqlExtDatabase(None)

    class Machine(peewee.Model):
        name = peewee.CharField(index=True, unique=True)
        dataset = BinaryJSONField()
        model = BinaryJSONField()
        metadata = BinaryJSONField()

        class Meta:
            database = db
            table_name = "machine"

        def __repr__(self):
            return f"Machine {self.__data__} "


class PostgresReporter(BaseReporter):
    """Reporter storing a Machine into Postgres."""

    db = db if HAS_PEEWEE else None

    @capture_args
    def __init__(
        self,
        host: str,
        port: int = 5432,
        user: str = "postgres",
        password: Optional[str] = None,
        database: str = "postgres",
        **kwargs,
    ):
        self.host = host
        self.port = port
        self.user = user
        self.password = password
        self.database = database
        if not HAS_PEEWEE:
            logger.warning(
                "peewee is not installed; PostgresReporter.report() will raise"
            )
            return
        sql_parameters = {"host": host, "port": port, "user": user}
        if password:
            sql_parameters["password"] = password
        sql_parameters.update(kwargs)
        self.db.init(database, **sql_parameters)

    def report(self, machine):
        if not HAS_PEEWEE:
            raise PostgresReporterException(
                "peewee is not installed in this environment"
            )
        try:
            with self.db.atomic():
                logger.info("Inserting machine %s in sql", machine.name)
                machine_dict = machine.to_dict()
                record = dict(
                    name=machine_dict["name"],
                    dataset=machine_dict["dataset"],
                    model=machine_dict["model"],
                    metadata=json.loads(machine.to_json())["metadata"],
                )
                Machine.insert(record).on_conflict(
                    conflict_target=[Machine.name],
                    update=record,
                ).execute()
        except Exception as exc:
            raise PostgresReporterException(exc) from exc
