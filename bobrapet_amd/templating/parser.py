"""Template + expression parser.

The template language is ``{{ expression }}`` segments embedded in strings,
evaluated over the scope ``{inputs, steps, run, story, branch, item}``.
Expressions are a CEL-like subset (the reference delegates to an external
templating engine combining Go templates + CEL — SURVEY.md §2.6; this is a
from-scratch equivalent with one unified grammar).

Grammar (precedence climbing):
    ternary   := or ('?' ternary ':' ternary)?
    or        := and ('||' and)*
    and       := not ('&&' not)*
    not       := '!' not | cmp
    cmp       := add (('=='|'!='|'<'|'<='|'>'|'>='|'in') add)?
    add       := mul (('+'|'-') mul)*
    mul       := unary (('*'|'/'|'%') unary)*
    unary     := '-' unary | postfix
    postfix   := primary ('.' IDENT | '[' ternary ']' | '(' args ')')*
    primary   := NUMBER | STRING | 'true' | 'false' | 'null' | IDENT
               | '.' IDENT postfix*          # Go-template-style leading dot
               | '(' ternary ')' | '[' args ']' | '{' kv-pairs '}'

Compiled ASTs are plain tuples: ('const', v), ('var', name), ('get', obj, key),
('index', obj, expr), ('call', fnname, [args]), ('method', obj, name, [args]),
('and', a, b), ('or', a, b), ('not', a), ('cmp', op, a, b), ('bin', op, a, b),
('neg', a), ('cond', c, a, b), ('list', [items]), ('map', [(k, v)]).
"""
from __future__ import annotations

import re
import typing as _t

TEMPLATE_RE = re.compile(r"\{\{(.*?)\}\}", re.S)


class TemplateSyntaxError(ValueError):
    pass


_TOKEN_RE = re.compile(
    r"""
    (?P<ws>\s+)
  | (?P<num>\d+\.\d+|\d+)
  | (?P<str>'(?:[^'\\]|\\.)*'|"(?:[^"\\]|\\.)*")
  | (?P<op>\|\||&&|==|!=|<=|>=|[<>+\-*/%!?:().,\[\]{}])
  | (?P<ident>[A-Za-z_][A-Za-z0-9_\-]*)
""",
    re.X,
)

_KEYWORDS = {"true": True, "false": False, "null": None, "none": None}


def _tokenize(src: str) -> _t.List[_t.Tuple[str, str]]:
    tokens = []
    pos = 0
    while pos < len(src):
        m = _TOKEN_RE.match(src, pos)
        if not m:
            raise TemplateSyntaxError(f"unexpected character {src[pos]!r} in {src!r}")
        pos = m.end()
        kind = m.lastgroup
        if kind == "ws":
            continue
        tokens.append((kind, m.group()))
    tokens.append(("eof", ""))
    return tokens


class _Parser:
    def __init__(self, src: str):
        self.src = src
        self.toks = _tokenize(src)
        self.i = 0

    def peek(self) -> _t.Tuple[str, str]:
        return self.toks[self.i]

    def next(self) -> _t.Tuple[str, str]:
        t = self.toks[self.i]
        self.i += 1
        return t

    def expect(self, text: str) -> None:
        kind, val = self.next()
        if val != text:
            raise TemplateSyntaxError(f"expected {text!r}, got {val!r} in {self.src!r}")

    def parse(self):
        node = self.ternary()
        if self.peek()[0] != "eof":
            raise TemplateSyntaxError(
                f"trailing tokens at {self.peek()[1]!r} in {self.src!r}"
            )
        return node

    def ternary(self):
        cond = self.or_()
        if self.peek()[1] == "?":
            self.next()
            a = self.ternary()
            self.expect(":")
            b = self.ternary()
            return ("cond", cond, a, b)
        return cond

    def or_(self):
        node = self.and_()
        while self.peek()[1] == "||":
            self.next()
            node = ("or", node, self.and_())
        return node

    def and_(self):
        node = self.not_()
        while self.peek()[1] == "&&":
            self.next()
            node = ("and", node, self.not_())
        return node

    def not_(self):
        if self.peek()[1] == "!":
            self.next()
            return ("not", self.not_())
        return self.cmp()

    def cmp(self):
        node = self.add()
        kind, val = self.peek()
        if val in ("==", "!=", "<", "<=", ">", ">=") or (kind == "ident" and val == "in"):
            self.next()
            return ("cmp", val, node, self.add())
        return node

    def add(self):
        node = self.mul()
        while self.peek()[1] in ("+", "-"):
            op = self.next()[1]
            node = ("bin", op, node, self.mul())
        return node

    def mul(self):
        node = self.unary()
        while self.peek()[1] in ("*", "/", "%"):
            op = self.next()[1]
            node = ("bin", op, node, self.unary())
        return node

    def unary(self):
        if self.peek()[1] == "-":
            self.next()
            return ("neg", self.unary())
        return self.postfix()

    def postfix(self, node=None):
        if node is None:
            node = self.primary()
        while True:
            kind, val = self.peek()
            if val == ".":
                self.next()
                k, name = self.next()
                if k not in ("ident", "num"):
                    raise TemplateSyntaxError(f"expected name after '.' in {self.src!r}")
                if self.peek()[1] == "(":
                    args = self._args()
                    node = ("method", node, name, args)
                else:
                    node = ("get", node, name)
            elif val == "[":
                self.next()
                idx = self.ternary()
                self.expect("]")
                node = ("index", node, idx)
            elif val == "(" and node[0] == "var":
                args = self._args()
                node = ("call", node[1], args)
            else:
                return node

    def _args(self):
        self.expect("(")
        args = []
        if self.peek()[1] != ")":
            args.append(self.ternary())
            while self.peek()[1] == ",":
                self.next()
                args.append(self.ternary())
        self.expect(")")
        return args

    def primary(self):
        kind, val = self.next()
        if kind == "num":
            return ("const", float(val) if "." in val else int(val))
        if kind == "str":
            body = val[1:-1]
            body = re.sub(r"\\(.)", lambda m: {"n": "\n", "t": "\t"}.get(m.group(1), m.group(1)), body)
            return ("const", body)
        if kind == "ident":
            if val in _KEYWORDS:
                return ("const", _KEYWORDS[val])
            return ("var", val)
        if val == "(":
            node = self.ternary()
            self.expect(")")
            return node
        if val == ".":
            # Go-template-style leading dot: `.steps.foo` == `steps.foo`
            k, name = self.next()
            if k != "ident":
                raise TemplateSyntaxError(f"expected name after leading '.' in {self.src!r}")
            return ("var", name)
        if val == "[":
            items = []
            if self.peek()[1] != "]":
                items.append(self.ternary())
                while self.peek()[1] == ",":
                    self.next()
                    items.append(self.ternary())
            self.expect("]")
            return ("list", items)
        if val == "{":
            pairs = []
            if self.peek()[1] != "}":
                pairs.append(self._kv())
                while self.peek()[1] == ",":
                    self.next()
                    pairs.append(self._kv())
            self.expect("}")
            return ("map", pairs)
        raise TemplateSyntaxError(f"unexpected token {val!r} in {self.src!r}")

    def _kv(self):
        kind, key = self.next()
        if kind == "str":
            key_node = ("const", key[1:-1])
        elif kind == "ident":
            key_node = ("const", key)
        else:
            raise TemplateSyntaxError(f"bad map key {key!r} in {self.src!r}")
        self.expect(":")
        return (key_node, self.ternary())


def parse_expression(src: str):
    """Parse one expression to an AST."""
    return _Parser(src).parse()


class CompiledTemplate:
    """A parsed template string: literal segments + expression ASTs."""

    __slots__ = ("source", "parts", "single")

    def __init__(self, source: str, parts, single: bool):
        self.source = source
        self.parts = parts  # list of ('lit', str) | ('expr', ast)
        self.single = single  # whole string is one expression → preserve type

    @property
    def is_static(self) -> bool:
        return all(kind == "lit" for kind, _ in self.parts)


def parse_template(text: str) -> CompiledTemplate:
    """Split a string into literal and {{ }} expression segments."""
    parts = []
    pos = 0
    for m in TEMPLATE_RE.finditer(text):
        if m.start() > pos:
            parts.append(("lit", text[pos : m.start()]))
        parts.append(("expr", parse_expression(m.group(1).strip())))
        pos = m.end()
    if pos < len(text):
        parts.append(("lit", text[pos:]))
    if not parts:
        parts = [("lit", text)]
    single = len(parts) == 1 and parts[0][0] == "expr"
    return CompiledTemplate(text, parts, single)


def is_template(text) -> bool:
    return isinstance(text, str) and "{{" in text and "}}" in text
