"""
Exception types of the in-repo data layer.

The reference imports these from the external ``gordo_core`` package
(reference: gordo/cli/cli.py:9-11); here they live in-repo.
"""


class ConfigException(ValueError):
    """Malformed dataset / machine configuration."""


class InsufficientDataError(ValueError):
    """Raised when a dataset yields too few rows to train on."""


class SensorTagNormalizationError(ValueError):
    """Raised when a sensor tag cannot be normalized."""


class NoSuitableDataProviderError(ValueError):
    """Raised when no data provider can serve the requested tags."""
