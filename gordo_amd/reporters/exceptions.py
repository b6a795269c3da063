"""Reporter exceptions (spec: gordo/reporters/exceptions.py)."""


class ReporterException(Exception):
    pass
