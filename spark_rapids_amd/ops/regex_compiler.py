"""Regex -> bytecode compiler for the GPU matcher (regex.hip).

Reference analogue: the reference transpiles Java regex to cudf's regex
dialect (RegexParser.scala, 2.1k LoC) and runs cudf's device regex engine.
Here the engine owns both halves: this compiler lowers a supported subset
to a compact backtracking-VM program; unsupported syntax raises
RegexUnsupported and the overrides pass keeps the expression on CPU
(python `re`).

Supported: literals, '.', escapes (\\d \\D \\w \\W \\s \\S \\. etc),
character classes [a-z0-9_] with negation and ranges, quantifiers * + ?
and {m}/{m,}/{m,n} (expanded, capped), non-capturing-style groups (...)
(treated as grouping only), alternation |, anchors ^ $.
Unsupported -> CPU: backreferences, lookaround, lazy quantifiers,
named groups, unicode classes.
"""
from __future__ import annotations

from typing import List, Tuple

OP_CHAR = 0    # arg0 = byte
OP_ANY = 1     # any byte except \n (java '.' default)
OP_CLASS = 2   # arg0 = class index, arg1 = negated
OP_MATCH = 3
OP_JMP = 4     # arg0 = target
OP_SPLIT = 5   # arg0, arg1 = targets (try arg0 first)
OP_BOL = 6
OP_EOL = 7
OP_SAVE = 8    # arg0 = save slot (2*group / 2*group+1): capture positions

MAX_EXPANSION = 32


class RegexUnsupported(ValueError):
    pass


class _Class:
    def __init__(self):
        self.bitmap = bytearray(32)  # 256 bits

    def add(self, b: int):
        self.bitmap[b >> 3] |= 1 << (b & 7)

    def add_range(self, lo: int, hi: int):
        for b in range(lo, hi + 1):
            self.add(b)


_ESCAPE_CLASSES = {
    "d": ("0", "9", False),
    "w": (None, None, False),
    "s": (None, None, False),
}


def _escape_class(ch: str) -> Tuple[_Class, bool]:
    c = _Class()
    neg = ch.isupper()
    base = ch.lower()
    if base == "d":
        c.add_range(ord("0"), ord("9"))
    elif base == "w":
        c.add_range(ord("a"), ord("z"))
        c.add_range(ord("A"), ord("Z"))
        c.add_range(ord("0"), ord("9"))
        c.add(ord("_"))
    elif base == "s":
        for b in b" \t\n\r\f\v":
            c.add(b)
    else:
        raise RegexUnsupported(f"escape \\{ch}")
    return c, neg


class Program:
    def __init__(self):
        self.ops: List[Tuple[int, int, int]] = []
        self.classes: List[bytes] = []
        self.anchored_start = False
        self.ngroups = 0  # capturing groups (slots 2..2*ngroups+1)

    def emit(self, op, a0=0, a1=0) -> int:
        self.ops.append((op, a0, a1))
        return len(self.ops) - 1

    def patch(self, idx, a0=None, a1=None):
        op, x, y = self.ops[idx]
        self.ops[idx] = (op, x if a0 is None else a0, y if a1 is None else a1)


class _Parser:
    def __init__(self, pattern: str):
        self.p = pattern
        self.i = 0
        self.prog = Program()

    def peek(self):
        return self.p[self.i] if self.i < len(self.p) else None

    def next(self):
        ch = self.p[self.i]
        self.i += 1
        return ch

    # fragment = list of op tuples (relative jumps resolved at append time);
    # we compile directly into the program using absolute indexes, returning
    # (start, ends) is complex — instead compile each atom into a sub-list
    # and concatenate, resolving jumps relative to the fragment start.
    def parse(self) -> Program:
        frag = self._alternation()
        frag.append((OP_MATCH, 0, 0))
        self.prog.ops = frag
        if self.i != len(self.p):
            raise RegexUnsupported(f"trailing at {self.p[self.i:]}")
        return self.prog

    def _alternation(self) -> list:
        branches = [self._sequence()]
        while self.peek() == "|":
            self.next()
            branches.append(self._sequence())
        if len(branches) == 1:
            return branches[0]
        # chain: SPLIT b1, next; b1; JMP end; ...
        out: list = []
        total = 0
        frags = branches
        # compute layout
        result: list = []
        end_jumps = []
        for k, f in enumerate(frags):
            if k < len(frags) - 1:
                split_at = len(result)
                result.append([OP_SPLIT, None, None])  # placeholder
                body_start = len(result)
                result.extend(self._shift(f, body_start))
                jmp_at = len(result)
                result.append([OP_JMP, None, 0])
                end_jumps.append(jmp_at)
                result[split_at][1] = body_start
                result[split_at][2] = len(result)
            else:
                start = len(result)
                result.extend(self._shift(f, start))
        end = len(result)
        for j in end_jumps:
            result[j][1] = end
        return [tuple(x) for x in result]

    def _shift(self, frag: list, base: int) -> list:
        out = []
        for op, a0, a1 in frag:
            if op in (OP_JMP,):
                out.append([op, a0 + base, a1])
            elif op == OP_SPLIT:
                out.append([op, a0 + base, a1 + base])
            else:
                out.append([op, a0, a1])
        return out

    def _sequence(self) -> list:
        frag: list = []
        while True:
            ch = self.peek()
            if ch is None or ch in "|)":
                return frag
            atom = self._atom()
            atom = self._quantify(atom)
            frag = frag + self._shift(atom, len(frag))

    def _atom(self) -> list:
        ch = self.next()
        if ch == "(":
            capturing = True
            if self.peek() == "?":
                # (?: ... ) non-capturing; anything else unsupported
                self.next()
                if self.peek() != ":":
                    raise RegexUnsupported("lookaround / named group")
                self.next()
                capturing = False
            gid = 0
            if capturing:
                self.prog.ngroups += 1
                gid = self.prog.ngroups
            inner = self._alternation()
            if self.peek() != ")":
                raise RegexUnsupported("unbalanced (")
            self.next()
            if capturing:
                inner = [(OP_SAVE, 2 * gid, 0)] + \
                    self._shift(inner, 1) + [(OP_SAVE, 2 * gid + 1, 0)]
            return inner
        if ch == "[":
            return [self._char_class()]
        if ch == ".":
            return [(OP_ANY, 0, 0)]
        if ch == "^":
            return [(OP_BOL, 0, 0)]
        if ch == "$":
            return [(OP_EOL, 0, 0)]
        if ch == "\\":
            e = self.next()
            if e.lower() in ("d", "w", "s"):
                c, neg = _escape_class(e)
                self.prog.classes.append(bytes(c.bitmap))
                return [(OP_CLASS, len(self.prog.classes) - 1, int(neg))]
            if e in ".\\+*?()[]{}|^$":
                return [(OP_CHAR, ord(e), 0)]
            if e == "n":
                return [(OP_CHAR, 10, 0)]
            if e == "t":
                return [(OP_CHAR, 9, 0)]
            raise RegexUnsupported(f"escape \\{e}")
        if ch in "*+?":
            raise RegexUnsupported(f"dangling {ch}")
        b = ch.encode("utf-8")
        if len(b) == 1:
            return [(OP_CHAR, b[0], 0)]
        # multi-byte utf-8 literal: sequence of byte matches
        return [(OP_CHAR, x, 0) for x in b]

    def _char_class(self):
        c = _Class()
        neg = False
        if self.peek() == "^":
            neg = True
            self.next()
        first = True
        while True:
            ch = self.peek()
            if ch is None:
                raise RegexUnsupported("unbalanced [")
            if ch == "]" and not first:
                self.next()
                break
            first = False
            self.next()
            if ch == "\\":
                e = self.next()
                if e.lower() in ("d", "w", "s"):
                    sub, sneg = _escape_class(e)
                    if sneg:
                        raise RegexUnsupported("negated escape in class")
                    for b in range(256):
                        if sub.bitmap[b >> 3] & (1 << (b & 7)):
                            c.add(b)
                    continue
                ch = {"n": "\n", "t": "\t"}.get(e, e)
            if self.peek() == "-" and self.i + 1 < len(self.p) \
                    and self.p[self.i + 1] != "]":
                self.next()
                hi = self.next()
                c.add_range(ord(ch), ord(hi))
            else:
                for b in ch.encode("utf-8"):
                    c.add(b)
        self.prog.classes.append(bytes(c.bitmap))
        return (OP_CLASS, len(self.prog.classes) - 1, int(neg))

    def _quantify(self, atom: list) -> list:
        ch = self.peek()
        if ch not in ("*", "+", "?", "{"):
            return atom
        if ch == "{":
            # {m} {m,} {m,n}
            j = self.p.find("}", self.i)
            if j < 0:
                raise RegexUnsupported("unbalanced {")
            spec = self.p[self.i + 1: j]
            self.i = j + 1
            if self.peek() == "?":
                raise RegexUnsupported("lazy quantifier")
            parts = spec.split(",")
            try:
                m = int(parts[0])
                n = int(parts[1]) if len(parts) > 1 and parts[1] else None
                unbounded = len(parts) > 1 and not parts[1]
            except ValueError as e:
                raise RegexUnsupported(f"bad quantifier {{{spec}}}") from e
            if m > MAX_EXPANSION or (n is not None and n > MAX_EXPANSION):
                raise RegexUnsupported("quantifier bound too large")
            out: list = []
            for _ in range(m):
                out = out + self._shift(atom, len(out))
            if unbounded:
                out = out + self._shift(self._star(atom), len(out))
            elif n is not None:
                for _ in range(n - m):
                    out = out + self._shift(self._opt(atom), len(out))
            return out
        self.next()
        if self.peek() == "?":
            raise RegexUnsupported("lazy quantifier")
        if ch == "*":
            return self._star(atom)
        if ch == "+":
            # atom; SPLIT 0, end
            out = [list(x) for x in atom]
            out.append([OP_SPLIT, 0, len(atom) + 1])
            return [tuple(x) for x in out]
        # ?
        return self._opt(atom)

    def _star(self, atom: list) -> list:
        # 0: SPLIT 1, end; 1..k: atom; k+1: JMP 0; end:
        out = [[OP_SPLIT, 1, len(atom) + 2]]
        out.extend(self._shift(atom, 1))
        out.append([OP_JMP, 0, 0])
        return [tuple(x) for x in out]

    def _opt(self, atom: list) -> list:
        out = [[OP_SPLIT, 1, len(atom) + 1]]
        out.extend(self._shift(atom, 1))
        return [tuple(x) for x in out]


def compile_regex(pattern: str) -> Program:
    """Compile to VM bytecode; raises RegexUnsupported outside the subset."""
    return _Parser(pattern).parse()
