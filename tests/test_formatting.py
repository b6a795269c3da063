"""Code hygiene enforced as a test (the reference runs black --check as
a test — tests/gordo/test_formatting.py there; black isn't in this
image, so this enforces the mechanically-checkable subset: every file
parses, no tabs in indentation, no trailing whitespace, and no unused
imports outside __init__ re-export modules)."""
import ast
import pathlib


PKG = pathlib.Path(__file__).resolve().parent.parent / "gordo_amd"


def _py_files():
    return sorted(p for p in PKG.rglob("*.py") if "__pycache__" not in str(p))


def test_all_files_parse():
    for p in _py_files():
        ast.parse(p.read_text(), filename=str(p))


def test_no_tabs_or_trailing_whitespace():
    offenders = []
    for p in _py_files():
        for i, line in enumerate(p.read_text().splitlines(), 1):
            if line.rstrip("\n") != line.rstrip():
                offenders.append(f"{p}:{i}: trailing whitespace")
            if line.startswith("\t"):
                offenders.append(f"{p}:{i}: tab indentation")
    assert not offenders, "\n".join(offenders[:40])


def test_no_unused_imports():
    offenders = []
    for p in _py_files():
        if p.name == "__init__.py":  # re-export modules
            continue
        src = p.read_text()
        tree = ast.parse(src)
        lines = src.splitlines()
        imported = {}
        for node in ast.walk(tree):
            if isinstance(node, ast.Import):
                for a in node.names:
                    imported[(a.asname or a.name).split(".")[0]] = node.lineno
            elif isinstance(node, ast.ImportFrom):
                for a in node.names:
                    if a.name != "*":
                        imported[a.asname or a.name] = node.lineno
        used = set()
        for node in ast.walk(tree):
            if isinstance(node, ast.Name):
                used.add(node.id)
            elif isinstance(node, ast.Attribute):
                n = node
                while isinstance(n, ast.Attribute):
                    n = n.value
                if isinstance(n, ast.Name):
                    used.add(n.id)
        for name, line in imported.items():
            if name in ("annotations",) or name in used:
                continue
            if "noqa" in lines[line - 1]:
                continue
            offenders.append(f"{p}:{line}: unused import {name}")
    assert not offenders, "\n".join(offenders)
