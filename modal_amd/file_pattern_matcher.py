"""FilePatternMatcher: dockerignore-style path filtering.

Parity: /root/reference/py/modal/file_pattern_matcher.py — used by
``Image.add_local_dir(ignore=...)`` and ``Mount`` conditions. Supports ``*``,
``**``, ``?``, character classes, and ``!`` negation, matched against
relative paths.
"""

from __future__ import annotations

import fnmatch
import os
from typing import Callable, Sequence, Union


class FilePatternMatcher:
    def __init__(self, *patterns: str):
        self._rules: list[tuple[bool, str]] = []
        for pattern in patterns:
            pattern = pattern.strip()
            if not pattern:
                continue
            negate = pattern.startswith("!")
            if negate:
                pattern = pattern[1:]
            self._rules.append((negate, pattern.strip("/")))

    @classmethod
    def from_file(cls, path: Union[str, os.PathLike]) -> "FilePatternMatcher":
        with open(path) as f:
            lines = [ln.strip() for ln in f if ln.strip() and not ln.startswith("#")]
        return cls(*lines)

    def _match_one(self, pattern: str, rel: str) -> bool:
        rel = rel.strip("/")
        if "**" in pattern:
            regex_parts = pattern.split("**")
            # fnmatch handles each side; '**' bridges any depth
            if pattern == "**":
                return True
            if pattern.startswith("**/"):
                tail = pattern[3:]
                return any(
                    fnmatch.fnmatch("/".join(rel.split("/")[i:]), tail)
                    for i in range(len(rel.split("/")))
                )
            if pattern.endswith("/**"):
                head = pattern[:-3]
                return fnmatch.fnmatch(rel, head) or rel.startswith(head + "/")
            head, _, tail = pattern.partition("/**/")
            parts = rel.split("/")
            for i in range(1, len(parts)):
                if fnmatch.fnmatch("/".join(parts[:i]), head) and fnmatch.fnmatch(
                    "/".join(parts[i:]), tail
                ):
                    return True
            return False
        if fnmatch.fnmatch(rel, pattern):
            return True
        # a directory pattern matches everything under it
        return rel.startswith(pattern + "/") or any(
            fnmatch.fnmatch(part, pattern) for part in rel.split("/")[:1]
        )

    def __call__(self, path: Union[str, os.PathLike]) -> bool:
        """True if the path matches the pattern set (respecting negations)."""
        rel = str(path).replace(os.sep, "/")
        matched = False
        for negate, pattern in self._rules:
            if self._match_one(pattern, rel):
                matched = not negate
        return matched


def _ignore_fn(ignore: Union[None, Sequence[str], Callable, FilePatternMatcher]) -> Callable[[str], bool]:
    if ignore is None:
        return lambda _p: False
    if callable(ignore):
        return ignore
    return FilePatternMatcher(*ignore)
