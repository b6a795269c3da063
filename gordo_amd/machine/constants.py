"""Machine-config fields whose YAML values may themselves be nested
YAML strings (``|`` blocks) — reference gordo/machine/constants.py:1-3."""

MACHINE_YAML_FIELDS = ("model", "dataset", "evaluation", "metadata", "runtime")
