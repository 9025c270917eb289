"""A Go ``text/template`` interpreter for the reference's templates.

Scope: the language surface actually used by
/root/reference/internal/plugins/workload/v1/scaffolds/templates/**
(see the package docstring).  ``with``, nested template definitions,
and ``block`` are deliberately unsupported — the reference never uses
them — and hitting one raises GoTemplateError so a gap is loud.

Data model: contexts are plain Python dicts / objects.  Field access
``.Name`` resolves a dict key or attribute; a callable resolved
mid-chain is invoked with no arguments (Go's niladic method calls);
a callable at the end of a command is invoked with the command's
arguments.  Go zero-value truthiness applies in ``if``/``and``/``or``.
"""

from __future__ import annotations

import re
from dataclasses import dataclass
from typing import Any, Callable, Optional


class GoTemplateError(Exception):
    pass


# ---- function library ---------------------------------------------------


def _go_true(v: Any) -> bool:
    """Go template truth: the zero value of any type is false."""
    if v is None or v is False:
        return False
    if isinstance(v, (int, float)) and not isinstance(v, bool) and v == 0:
        return False
    if isinstance(v, (str, list, tuple, dict)) and len(v) == 0:
        return False
    return True


def _fmt_value(v: Any) -> str:
    """Go's %v / default stringification for the types templates emit."""
    if isinstance(v, bool):
        return "true" if v else "false"
    if v is None:
        return "<nil>"
    if isinstance(v, float) and v == int(v):
        return str(int(v))
    return str(v)


def _printf(spec: str, *args: Any) -> str:
    out: list[str] = []
    i = 0
    ai = 0
    n = len(spec)
    while i < n:
        ch = spec[i]
        if ch != "%":
            out.append(ch)
            i += 1
            continue
        if i + 1 < n and spec[i + 1] == "%":
            out.append("%")
            i += 2
            continue
        m = re.match(r"%[-+ #0]*\d*(?:\.\d+)?[sdvqxXtf]", spec[i:])
        if not m:
            raise GoTemplateError(f"unsupported printf verb at {spec[i:]!r}")
        verb = m.group()[-1]
        arg = args[ai]
        ai += 1
        if verb == "q":
            out.append('"' + str(arg).replace("\\", "\\\\").replace('"', '\\"') + '"')
        elif verb in "dxX":
            fmtd = m.group().replace("v", "d")
            out.append(fmtd % int(arg))
        elif verb == "f":
            out.append(m.group() % float(arg))
        elif verb == "t":
            out.append("true" if arg else "false")
        else:  # s, v
            out.append(m.group().replace("v", "s").replace("t", "s") % _fmt_value(arg))
        i += len(m.group())
    return "".join(out)


def _hash_fnv(s: str) -> str:
    """kubebuilder machinery's hashFNV: fnv-1a 32-bit, hex."""
    h = 0x811C9DC5
    for b in s.encode("utf-8"):
        h ^= b
        h = (h * 0x01000193) & 0xFFFFFFFF
    return f"{h:08x}"


def _go_title(s: str) -> str:
    # strings.Title semantics (first letter of each word)
    return re.sub(
        r"\b\w", lambda m: m.group().upper(), s
    )


def _quote_string(value: str) -> str:
    # reference internal/utils/functionmap.go:21-33
    if not value.startswith('"'):
        value = '"' + value
    if not value.endswith('"'):
        value = value + '"'
    return value


BUILTINS: dict[str, Callable] = {
    "printf": _printf,
    "print": lambda *a: "".join(_fmt_value(x) for x in a),
    "len": lambda v: len(v),
    "not": lambda v: not _go_true(v),
    "eq": lambda a, *rest: any(a == r for r in rest),
    "ne": lambda a, b: a != b,
    "lt": lambda a, b: a < b,
    "le": lambda a, b: a <= b,
    "gt": lambda a, b: a > b,
    "ge": lambda a, b: a >= b,
    "index": lambda v, *keys: _index(v, keys),
    # kubebuilder machinery DefaultFuncMap
    "lower": lambda s: s.lower(),
    "title": _go_title,
    "isEmptyStr": lambda s: s == "",
    "hashFNV": _hash_fnv,
    # reference internal/utils/functionmap.go
    "quoteString": _quote_string,
    "removeString": lambda value, with_: with_.replace(value, ""),
    "containsString": lambda value, in_: value in in_,
}


def _index(v: Any, keys: tuple) -> Any:
    for k in keys:
        v = v[k]
    return v


# ---- parsing ------------------------------------------------------------


@dataclass
class _Text:
    text: str


@dataclass
class _Output:
    pipe: "_Pipe"


@dataclass
class _If:
    # list of (condition-pipe or None for else, body)
    branches: list

@dataclass
class _Range:
    key_var: Optional[str]
    val_var: Optional[str]
    pipe: "_Pipe"
    body: list
    else_body: list


@dataclass
class _VarSet:
    name: str
    pipe: "_Pipe"
    declare: bool


# pipeline AST -------------------------------------------------------------


@dataclass
class _Pipe:
    commands: list  # list of _Command


@dataclass
class _Command:
    operand: Any  # _Chain | _Var | _Lit | _Paren | str (function name)
    args: list


@dataclass
class _Chain:
    base: Any  # None for '.', _Var, or _Paren
    fields: list


@dataclass
class _Var:
    name: str  # without '$'; '' means the root var $


@dataclass
class _Lit:
    value: Any


@dataclass
class _Paren:
    pipe: _Pipe


_ACTION_RE = re.compile(r"{{(-)?\s*(.*?)\s*(-)?}}", re.S)

_TOKEN_RE = re.compile(
    r"""
      (?P<str>"(?:\\.|[^"\\])*")
    | (?P<raw>`[^`]*`)
    | (?P<num>-?\d+(?:\.\d+)?)
    | (?P<decl>:=|=)
    | (?P<pipe>\|)
    | (?P<lparen>\()
    | (?P<rparen>\))
    | (?P<comma>,)
    | (?P<chain>[.$][\w.$]*)
    | (?P<ident>\w+)
    """,
    re.VERBOSE,
)


def _lex_action(src: str) -> list[tuple[str, str]]:
    tokens: list[tuple[str, str]] = []
    pos = 0
    while pos < len(src):
        if src[pos].isspace():
            pos += 1
            continue
        m = _TOKEN_RE.match(src, pos)
        if not m:
            raise GoTemplateError(f"bad token in action: {src[pos:]!r}")
        tokens.append((m.lastgroup, m.group()))
        pos = m.end()
    return tokens


def _parse_operand(tokens: list, i: int):
    kind, text = tokens[i]
    if kind == "str":
        # Go escape subset
        s = text[1:-1]
        s = (
            s.replace("\\n", "\n")
            .replace("\\t", "\t")
            .replace('\\"', '"')
            .replace("\\\\", "\\")
        )
        return _Lit(s), i + 1
    if kind == "raw":
        return _Lit(text[1:-1]), i + 1
    if kind == "num":
        v = float(text) if "." in text else int(text)
        return _Lit(v), i + 1
    if kind == "lparen":
        pipe, i = _parse_pipe(tokens, i + 1)
        if i >= len(tokens) or tokens[i][0] != "rparen":
            raise GoTemplateError("missing )")
        return _Paren(pipe), i + 1
    if kind == "chain":
        return _parse_chain(text), i + 1
    if kind == "ident":
        if text == "true":
            return _Lit(True), i + 1
        if text == "false":
            return _Lit(False), i + 1
        if text == "nil":
            return _Lit(None), i + 1
        return text, i + 1  # function name
    raise GoTemplateError(f"unexpected token {text!r}")


def _parse_chain(text: str):
    if text.startswith("$"):
        # $, $x, $x.F, $.F
        rest = text[1:]
        if rest.startswith("."):
            return _Chain(_Var(""), [f for f in rest[1:].split(".") if f])
        if "." in rest:
            name, _, fields = rest.partition(".")
            return _Chain(_Var(name), fields.split("."))
        return _Var(rest) if rest else _Var("")
    # .A.B / plain .
    fields = [f for f in text.split(".") if f]
    return _Chain(None, fields)


def _parse_command(tokens: list, i: int):
    operand, i = _parse_operand(tokens, i)
    args = []
    while i < len(tokens) and tokens[i][0] not in ("pipe", "rparen"):
        arg, i = _parse_operand(tokens, i)
        args.append(arg)
    return _Command(operand, args), i


def _parse_pipe(tokens: list, i: int):
    commands = []
    cmd, i = _parse_command(tokens, i)
    commands.append(cmd)
    while i < len(tokens) and tokens[i][0] == "pipe":
        cmd, i = _parse_command(tokens, i + 1)
        commands.append(cmd)
    return _Pipe(commands), i


def _parse_action(src: str):
    """Parse one action's interior into a control token or node."""
    tokens = _lex_action(src)
    if not tokens:
        raise GoTemplateError("empty action")

    kind, text = tokens[0]
    if kind == "ident" and text in ("if", "range", "else", "end", "with",
                                    "template", "block", "define"):
        if text in ("with", "template", "block", "define"):
            raise GoTemplateError(f"unsupported action {text!r}")
        if text == "end":
            return ("end", None)
        if text == "else":
            if len(tokens) > 1:
                if tokens[1][1] != "if":
                    raise GoTemplateError("expected `else if`")
                pipe, i = _parse_pipe(tokens, 2)
                return ("elseif", pipe)
            return ("else", None)
        if text == "if":
            pipe, i = _parse_pipe(tokens, 1)
            return ("if", pipe)
        # range [$k[, $v] :=] pipeline
        i = 1
        key_var = val_var = None
        if (
            tokens[i][0] == "chain"
            and tokens[i][1].startswith("$")
        ) and any(t[0] == "decl" for t in tokens):
            key_var = tokens[i][1][1:]
            i += 1
            if tokens[i][0] == "comma":
                val_var = tokens[i + 1][1][1:]
                i += 2
            if tokens[i][0] != "decl":
                raise GoTemplateError("expected := in range")
            i += 1
        pipe, i = _parse_pipe(tokens, i)
        return ("range", (key_var, val_var, pipe))

    # $x := pipe / $x = pipe
    if (
        kind == "chain"
        and text.startswith("$")
        and "." not in text
        and len(tokens) > 1
        and tokens[1][0] == "decl"
    ):
        pipe, i = _parse_pipe(tokens, 2)
        return ("var", _VarSet(text[1:], pipe, tokens[1][1] == ":="))

    pipe, i = _parse_pipe(tokens, 0)
    if i != len(tokens):
        raise GoTemplateError(f"trailing tokens in action {src!r}")
    return ("output", _Output(pipe))


def _parse_template(src: str) -> list:
    """Full template text -> node list (handles trim markers)."""
    nodes_stack: list[list] = [[]]
    control_stack: list[tuple] = []  # ("if", branches) | ("range", node)

    pos = 0
    pending_trim = False
    for m in _ACTION_RE.finditer(src):
        text = src[pos : m.start()]
        if pending_trim:
            text = text.lstrip(" \t\n\r")
        if m.group(1):  # {{- : trim whitespace before
            text = text.rstrip(" \t\n\r")
        if text:
            nodes_stack[-1].append(_Text(text))
        pending_trim = bool(m.group(3))
        pos = m.end()

        body = m.group(2)
        if body.startswith("/*"):
            continue  # comment

        kind, payload = _parse_action(body)

        if kind == "output":
            nodes_stack[-1].append(payload)
        elif kind == "var":
            nodes_stack[-1].append(payload)
        elif kind == "if":
            node = _If(branches=[(payload, [])])
            nodes_stack[-1].append(node)
            control_stack.append(("if", node))
            nodes_stack.append(node.branches[0][1])
        elif kind == "elseif":
            if not control_stack or control_stack[-1][0] != "if":
                raise GoTemplateError("else if outside if")
            node = control_stack[-1][1]
            nodes_stack.pop()
            node.branches.append((payload, []))
            nodes_stack.append(node.branches[-1][1])
        elif kind == "else":
            if not control_stack:
                raise GoTemplateError("else outside control")
            ckind, node = control_stack[-1]
            nodes_stack.pop()
            if ckind == "if":
                node.branches.append((None, []))
                nodes_stack.append(node.branches[-1][1])
            else:
                nodes_stack.append(node.else_body)
        elif kind == "range":
            key_var, val_var, pipe = payload
            node = _Range(key_var, val_var, pipe, [], [])
            nodes_stack[-1].append(node)
            control_stack.append(("range", node))
            nodes_stack.append(node.body)
        elif kind == "end":
            if not control_stack:
                raise GoTemplateError("end without open control")
            control_stack.pop()
            nodes_stack.pop()

    text = src[pos:]
    if pending_trim:
        text = text.lstrip(" \t\n\r")
    if text:
        nodes_stack[-1].append(_Text(text))

    if control_stack:
        raise GoTemplateError("unclosed control structure")
    return nodes_stack[0]


# ---- evaluation ----------------------------------------------------------


class _Scope:
    def __init__(self, root: Any):
        self.vars: dict[str, Any] = {"": root}
        self.stack: list[dict] = []

    def get(self, name: str) -> Any:
        for frame in reversed(self.stack):
            if name in frame:
                return frame[name]
        if name in self.vars:
            return self.vars[name]
        raise GoTemplateError(f"undefined variable ${name}")

    def set(self, name: str, value: Any, declare: bool) -> None:
        if declare:
            (self.stack[-1] if self.stack else self.vars)[name] = value
            return
        for frame in reversed(self.stack):
            if name in frame:
                frame[name] = value
                return
        if name in self.vars:
            self.vars[name] = value
            return
        raise GoTemplateError(f"assignment to undeclared ${name}")


class GoTemplate:
    def __init__(self, src: str, funcs: Optional[dict] = None):
        self.nodes = _parse_template(src)
        self.funcs = dict(BUILTINS)
        if funcs:
            self.funcs.update(funcs)

    def render(self, context: Any) -> str:
        out: list[str] = []
        scope = _Scope(context)
        self._exec(self.nodes, context, scope, out)
        return "".join(out)

    # -- execution

    def _exec(self, nodes: list, dot: Any, scope: _Scope, out: list) -> None:
        for node in nodes:
            if isinstance(node, _Text):
                out.append(node.text)
            elif isinstance(node, _Output):
                out.append(_fmt_value(self._pipe(node.pipe, dot, scope)))
            elif isinstance(node, _VarSet):
                scope.set(
                    node.name,
                    self._pipe(node.pipe, dot, scope),
                    node.declare,
                )
            elif isinstance(node, _If):
                for cond, body in node.branches:
                    if cond is None or _go_true(
                        self._pipe(cond, dot, scope)
                    ):
                        self._exec(body, dot, scope, out)
                        break
            elif isinstance(node, _Range):
                self._range(node, dot, scope, out)
            else:
                raise GoTemplateError(f"unknown node {node!r}")

    def _range(self, node: _Range, dot: Any, scope: _Scope, out: list):
        value = self._pipe(node.pipe, dot, scope)
        items: list[tuple[Any, Any]]
        if isinstance(value, dict):
            # text/template visits maps in sorted key order
            items = [(k, value[k]) for k in sorted(value)]
        elif value is None:
            items = []
        else:
            items = list(enumerate(value))

        if not items:
            self._exec(node.else_body, dot, scope, out)
            return

        scope.stack.append({})
        try:
            for k, v in items:
                if node.key_var is not None and node.val_var is None:
                    # `range $x := pipe` binds the VALUE
                    scope.stack[-1][node.key_var] = v
                else:
                    if node.key_var is not None:
                        scope.stack[-1][node.key_var] = k
                    if node.val_var is not None:
                        scope.stack[-1][node.val_var] = v
                self._exec(node.body, v, scope, out)
        finally:
            scope.stack.pop()

    # -- pipelines

    def _pipe(self, pipe: _Pipe, dot: Any, scope: _Scope) -> Any:
        value: Any = None
        have_value = False
        for cmd in pipe.commands:
            extra = [value] if have_value else []
            value = self._command(cmd, extra, dot, scope)
            have_value = True
        return value

    def _command(
        self, cmd: _Command, extra: list, dot: Any, scope: _Scope
    ) -> Any:
        args = [self._operand_value(a, dot, scope) for a in cmd.args]
        args += extra

        op = cmd.operand
        if isinstance(op, str):  # function
            if op in ("and", "or"):
                # short-circuit, return the deciding operand (Go spec)
                operands = cmd.args + ([_Lit(extra[0])] if extra else [])
                result = None
                for o in operands:
                    result = self._operand_value(o, dot, scope)
                    truth = _go_true(result)
                    if (op == "and" and not truth) or (
                        op == "or" and truth
                    ):
                        return result
                return result
            fn = self.funcs.get(op)
            if fn is None:
                raise GoTemplateError(f"undefined function {op!r}")
            return fn(*args)

        base = self._operand_value(op, dot, scope, call_args=args)
        return base

    def _operand_value(
        self, op: Any, dot: Any, scope: _Scope, call_args: list = None
    ) -> Any:
        if isinstance(op, _Lit):
            return op.value
        if isinstance(op, _Var):
            return scope.get(op.name)
        if isinstance(op, _Paren):
            return self._pipe(op.pipe, dot, scope)
        if isinstance(op, _Chain):
            if op.base is None:
                value = dot
            else:
                value = self._operand_value(op.base, dot, scope)
            fields = op.fields
            for idx, f in enumerate(fields):
                value = self._field(value, f)
                last = idx == len(fields) - 1
                if callable(value):
                    if last and call_args:
                        value = value(*call_args)
                    else:
                        value = value()
            if not fields and callable(value) and call_args:
                value = value(*call_args)
            return value
        if isinstance(op, str):
            # bare function used as an argument value (niladic)
            fn = self.funcs.get(op)
            if fn is None:
                raise GoTemplateError(f"undefined function {op!r}")
            return fn()
        raise GoTemplateError(f"bad operand {op!r}")

    @staticmethod
    def _field(value: Any, name: str) -> Any:
        if isinstance(value, dict):
            if name in value:
                return value[name]
            raise GoTemplateError(f"missing field {name!r} in {list(value)[:8]}")
        attr = getattr(value, name, _SENTINEL)
        if attr is _SENTINEL:
            raise GoTemplateError(
                f"missing field {name!r} on {type(value).__name__}"
            )
        return attr


_SENTINEL = object()
