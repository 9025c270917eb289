"""Extract Go template bodies from the reference's template source.

Every reference template is a ``const``/``var`` raw-string literal
(backtick) in a readable .go file, e.g.::

    const typesTemplate = `{{ .Boilerplate }} ...`

Some templates are composed at runtime with ``fmt.Sprintf(body, args)``
(e.g. templates/main.go:33-38 substitutes scaffold-marker strings into
%s verbs); callers supply those argument values (with citations) and
``template_body`` applies the substitution.
"""

from __future__ import annotations

import re
from functools import lru_cache

# start of a string assignment: `name = ` followed by a raw or quoted
# string; the value may be a +-concatenation chain (the templates embed
# literal backticks as `...` + "`" + `...`)
_ASSIGN_START = re.compile(r"(\w+)\s*=\s*(?=[`\"])")
_RAW = re.compile(r"`([^`]*)`", re.S)
_QUOTED = re.compile(r'"((?:\\.|[^"\\])*)"')
_PLUS = re.compile(r"\s*\+\s*")

_ESCAPES = {"n": "\n", "t": "\t", '"': '"', "\\": "\\", "`": "`"}


def _unquote(s: str) -> str:
    out: list[str] = []
    i = 0
    while i < len(s):
        if s[i] == "\\" and i + 1 < len(s):
            out.append(_ESCAPES.get(s[i + 1], s[i + 1]))
            i += 2
        else:
            out.append(s[i])
            i += 1
    return "".join(out)


@lru_cache(maxsize=None)
def extract_raw_strings(path: str) -> dict[str, str]:
    """name -> full string value for every string assignment in the
    file, following +-concatenation chains of raw/quoted literals."""
    with open(path, encoding="utf-8") as f:
        src = f.read()

    out: dict[str, str] = {}
    for m in _ASSIGN_START.finditer(src):
        name = m.group(1)
        pos = m.end()
        parts: list[str] = []
        while True:
            rm = _RAW.match(src, pos)
            if rm:
                parts.append(rm.group(1))
                pos = rm.end()
            else:
                qm = _QUOTED.match(src, pos)
                if not qm:
                    break
                parts.append(_unquote(qm.group(1)))
                pos = qm.end()
            pm = _PLUS.match(src, pos)
            if not pm:
                break
            pos = pm.end()
        if parts:
            # a name assigned more than once keeps its longest value
            # (template consts are unique; short ones are incidental)
            value = "".join(parts)
            if len(value) > len(out.get(name, "")):
                out[name] = value
    return out


def template_body(
    path: str, name: str, sprintf_args: tuple[str, ...] = ()
) -> str:
    """The template body named ``name`` in ``path``; if the reference
    composes it with fmt.Sprintf, pass the argument values in order."""
    bodies = extract_raw_strings(path)
    if name not in bodies:
        raise KeyError(
            f"no raw string {name!r} in {path} (have {sorted(bodies)})"
        )
    body = bodies[name]
    if sprintf_args:
        # Go %s/%v substitution only (what the references use)
        out: list[str] = []
        i = 0
        ai = 0
        while i < len(body):
            if body.startswith("%%", i):
                out.append("%")
                i += 2
            elif body.startswith("%s", i) or body.startswith("%v", i):
                out.append(str(sprintf_args[ai]))
                ai += 1
                i += 2
            else:
                out.append(body[i])
                i += 1
        if ai != len(sprintf_args):
            raise ValueError(
                f"{name}: {len(sprintf_args)} sprintf args given, "
                f"{ai} consumed"
            )
        body = "".join(out)
    return body
