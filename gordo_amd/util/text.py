"""Text helpers (spec: gordo/util/text.py)."""


def replace_all_non_ascii_chars(s: str, replacement: str = "_") -> str:
    """
    >>> replace_all_non_ascii_chars("héllo", "?")
    'h?llo'
    """
    return "".join(c if ord(c) < 128 else replacement for c in s)
