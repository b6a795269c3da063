"""Reuse the test suite's trained-model/server fixtures for the
benchmark lane (run explicitly: pytest benchmarks/ -q)."""
import importlib.util
import os

_path = os.path.join(os.path.dirname(__file__), "..", "tests", "conftest.py")
_spec = importlib.util.spec_from_file_location("gordo_test_fixtures", _path)
_mod = importlib.util.module_from_spec(_spec)
_spec.loader.exec_module(_mod)

globals().update(
    {k: v for k, v in vars(_mod).items() if not k.startswith("_")}
)
