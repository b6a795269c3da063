"""
Docker-image version-tag parsing (spec: gordo/util/version.py:88-130).
Drives imagePullPolicy selection in the workflow generator.
"""
from __future__ import annotations

import re
from dataclasses import dataclass
from typing import Optional


class Version:
    def get_version(self) -> str:
        raise NotImplementedError()


@dataclass(frozen=True)
class GordoRelease(Version):
    """A release tag like ``1.2.3`` / ``1.2`` / ``1``."""

    major: int
    minor: Optional[int] = None
    patch: Optional[int] = None
    suffix: Optional[str] = None

    def get_version(self) -> str:
        parts = [str(self.major)]
        if self.minor is not None:
            parts.append(str(self.minor))
        if self.patch is not None:
            parts.append(str(self.patch))
        return ".".join(parts) + (self.suffix or "")

    def only_major(self) -> bool:
        return self.minor is None and self.patch is None

    def only_major_minor(self) -> bool:
        return self.minor is not None and self.patch is None


@dataclass(frozen=True)
class GordoSpecial(Version):
    """A special tag: ``latest``, ``stable``."""

    special: str

    def get_version(self) -> str:
        return self.special


@dataclass(frozen=True)
class GordoPR(Version):
    """A PR tag: ``pr-123``."""

    number: int

    def get_version(self) -> str:
        return f"pr-{self.number}"


@dataclass(frozen=True)
class GordoSHA(Version):
    """A git-SHA tag."""

    sha: str

    def get_version(self) -> str:
        return self.sha


class Special:
    LATEST = "latest"
    STABLE = "stable"


_RELEASE_RE = re.compile(r"^(\d+)(?:\.(\d+))?(?:\.(\d+))?([\-.+][\w.\-+]+)?$")
_PR_RE = re.compile(r"^pr-(\d+)$")
_SHA_RE = re.compile(r"^[0-9a-f]{7,40}$")


def parse_version(version: str) -> Version:
    """
    >>> parse_version("1.2.3")
    GordoRelease(major=1, minor=2, patch=3, suffix=None)
    >>> parse_version("latest")
    GordoSpecial(special='latest')
    >>> parse_version("pr-42")
    GordoPR(number=42)
    """
    if version in (Special.LATEST, Special.STABLE):
        return GordoSpecial(version)
    m = _PR_RE.match(version)
    if m:
        return GordoPR(int(m.group(1)))
    m = _RELEASE_RE.match(version)
    if m:
        major, minor, patch, suffix = m.groups()
        return GordoRelease(
            int(major),
            int(minor) if minor is not None else None,
            int(patch) if patch is not None else None,
            suffix,
        )
    if _SHA_RE.match(version):
        return GordoSHA(version)
    raise ValueError(f"Unparsable version tag {version!r}")
