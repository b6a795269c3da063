"""
ExceptionsReporter — exception → exit-code mapping + JSON report file
(trimmed to the 2024-byte k8s termination-message limit).

Behavioral spec: gordo/cli/exceptions_reporter.py:35-221.
"""
from __future__ import annotations

import enum
import json
import traceback
from collections import Counter
from types import TracebackType
from typing import IO, Dict, Iterable, List, Optional, Tuple, Type

DEFAULT_EXIT_CODE = 1


class ReportLevel(enum.Enum):
    EXIT_CODE = 0
    TYPE = 1
    MESSAGE = 2
    TRACEBACK = 3

    @classmethod
    def get_names(cls) -> List[str]:
        return [level.name for level in cls]

    @classmethod
    def get_by_name(cls, name: str, default: Optional["ReportLevel"] = None):
        for level in cls:
            if level.name == name:
                return level
        return default


class ExceptionsReporter:
    """Save exception information as JSON (used as the k8s pod
    termination message by the builder)."""

    def __init__(
        self,
        exceptions: Iterable[Tuple[Type[Exception], int]],
        default_exit_code: int = DEFAULT_EXIT_CODE,
        traceback_limit: Optional[int] = None,
    ):
        self.exceptions_items = self.sort_exceptions(exceptions)
        self.default_exit_code = default_exit_code
        self.traceback_limit = traceback_limit

    @staticmethod
    def sort_exceptions(
        exceptions: Iterable[Tuple[Type[Exception], int]]
    ) -> List[Tuple[Type[Exception], int]]:
        """Sort so subclasses are found before their bases
        (inheritance-aware, reference :62-77)."""
        exceptions = list(exceptions)  # may be a generator; reused 3x
        inheritance_levels: Dict[Type[BaseException], int] = Counter()
        for exc, _ in exceptions:
            for e, _ in exceptions:
                if e is not exc and issubclass(exc, e):
                    inheritance_levels[e] += 1
        return sorted(
            list(exceptions),
            key=lambda v: (inheritance_levels[v[0]], v[1]),
        )

    @staticmethod
    def trim_message(message: str, max_length: int) -> str:
        if len(message) > max_length:
            message = message[: max_length - 3]
            return "" if len(message) <= 3 else message + "..."
        return message

    @staticmethod
    def trim_formatted_traceback(
        formatted_traceback: List[str], max_length: int
    ) -> List[str]:
        if sum(len(line) for line in formatted_traceback) <= max_length:
            return formatted_traceback
        length = 4
        result = []
        for line in reversed(formatted_traceback):
            length += len(line)
            if length > max_length:
                result.append("...\n")
                break
            result.append(line)
        return list(reversed(result))

    def found_exception_item(self, exc_type: Type[BaseException]):
        for item in self.exceptions_items:
            if issubclass(exc_type, item[0]):
                return item
        return None

    def exception_exit_code(self, exc_type: Optional[Type[BaseException]]) -> int:
        if exc_type is None:
            return 0
        item = self.found_exception_item(exc_type)
        return item[1] if item is not None else self.default_exit_code

    def report(
        self,
        level: ReportLevel,
        exc_type: Optional[Type[BaseException]],
        exc_value: Optional[BaseException],
        exc_traceback: Optional[TracebackType],
        report_file: IO[str],
        max_message_len: Optional[int] = None,
    ):
        report: Dict[str, object] = {}
        if exc_type is not None:
            report["exit_code"] = self.exception_exit_code(exc_type)
            if level.value >= ReportLevel.TYPE.value:
                report["type"] = exc_type.__name__
            if level.value >= ReportLevel.MESSAGE.value:
                message = str(exc_value) if exc_value is not None else ""
                if max_message_len is not None:
                    message = self.trim_message(message, max_message_len)
                report["message"] = message
            if level.value >= ReportLevel.TRACEBACK.value and exc_traceback:
                formatted = traceback.format_exception(
                    exc_type, exc_value, exc_traceback, limit=self.traceback_limit
                )
                if max_message_len is not None:
                    formatted = self.trim_formatted_traceback(
                        formatted, max_message_len
                    )
                report["traceback"] = "".join(formatted)
        else:
            report["exit_code"] = 0
        json.dump(report, report_file)

    def safe_report(
        self,
        level: ReportLevel,
        exc_type,
        exc_value,
        exc_traceback,
        report_file_path: str,
        max_message_len: Optional[int] = None,
    ):
        try:
            with open(report_file_path, "w") as report_file:
                self.report(
                    level, exc_type, exc_value, exc_traceback,
                    report_file, max_message_len,
                )
        except Exception:  # never let reporting mask the real error
            traceback.print_exc()
