"""Glob matching with Go `path.Match` semantics.

Oracle: infra/config/safety_policy.go:347-363 (topic patterns are matched with
path.Match; '*' does not cross '/', '?' matches one non-'/' rune, '[...]'
character classes with ranges and '^'/'!' negation, '\\' escapes).

Topics use '.' separators, which path.Match treats as ordinary characters, so
`job.*` matches both `job.default` and `job.a.b` — the evaluator preserves
that exact behavior. The device policy kernel pre-compiles patterns into
prefix/suffix/wildcard forms (ops/policy_compile.py); this host matcher is the
oracle for its tests and handles the general-pattern fallback.
"""
from __future__ import annotations


def glob_match(pattern: str, name: str) -> bool:
    """Go path.Match(pattern, name). Raises ValueError on malformed pattern."""
    return _match(pattern, 0, name, 0)


def _match(p: str, pi: int, s: str, si: int) -> bool:
    while pi < len(p):
        c = p[pi]
        if c == "*":
            # collapse consecutive stars
            while pi < len(p) and p[pi] == "*":
                pi += 1
            if pi == len(p):
                # trailing * matches rest if it has no '/'
                return "/" not in s[si:]
            # try to match remainder at every split point not crossing '/'
            for k in range(si, len(s) + 1):
                if _match(p, pi, s, k):
                    return True
                if k < len(s) and s[k] == "/":
                    return False
            return False
        elif c == "?":
            if si >= len(s) or s[si] == "/":
                return False
            pi += 1
            si += 1
        elif c == "[":
            if si >= len(s) or s[si] == "/":
                return False
            ok, pi = _match_class(p, pi, s[si])
            if not ok:
                return False
            si += 1
        elif c == "\\":
            pi += 1
            if pi >= len(p):
                raise ValueError("syntax error in pattern")
            if si >= len(s) or s[si] != p[pi]:
                return False
            pi += 1
            si += 1
        else:
            if si >= len(s) or s[si] != c:
                return False
            pi += 1
            si += 1
    return si == len(s)


def _match_class(p: str, pi: int, ch: str):
    # pi points at '['
    pi += 1
    negate = False
    if pi < len(p) and p[pi] in "^!":
        negate = True
        pi += 1
    matched = False
    first = True
    while True:
        if pi >= len(p):
            raise ValueError("syntax error in pattern")
        if p[pi] == "]" and not first:
            pi += 1
            break
        first = False
        lo = p[pi]
        if lo == "\\":
            pi += 1
            if pi >= len(p):
                raise ValueError("syntax error in pattern")
            lo = p[pi]
        pi += 1
        hi = lo
        if pi < len(p) and p[pi] == "-":
            pi += 1
            if pi >= len(p):
                raise ValueError("syntax error in pattern")
            hi = p[pi]
            if hi == "\\":
                pi += 1
                if pi >= len(p):
                    raise ValueError("syntax error in pattern")
                hi = p[pi]
            pi += 1
        if lo <= ch <= hi:
            matched = True
    return matched != negate, pi


def topic_matches(pattern: str, topic: str) -> bool:
    """Policy topic match: exact, or path.Match glob; malformed pattern -> no match."""
    if pattern == topic:
        return True
    try:
        return glob_match(pattern, topic)
    except ValueError:
        return False
