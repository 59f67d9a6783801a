"""SQL tokenizer.

Hand-written single-pass lexer for Spark SQL text (the reference's analogue is
a chumsky-based lexer, ref: crates/sail-sql-parser/src/lexer.rs). Produces a
flat token stream for the Pratt parser.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional


@dataclass
class Token:
    kind: str  # ident | number | string | op | eof
    value: str
    pos: int
    upper: str = ""

    def __post_init__(self):
        if self.kind == "ident":
            self.upper = self.value.upper()


from ..errors import ParseException


class SqlError(ParseException):
    def __init__(self, msg: str, sql: str = "", pos: int = -1):
        if pos >= 0 and sql:
            line = sql.count("\n", 0, pos) + 1
            col = pos - (sql.rfind("\n", 0, pos) + 1) + 1
            ctx = sql[max(0, pos - 30) : pos + 30].replace("\n", " ")
            msg = f"{msg} at line {line}:{col} near ...{ctx}..."
        super().__init__(msg)


_MULTI_OPS = ["<=>", "<>", "!=", ">=", "<=", "||", "::", "->", "=>"]
_SINGLE_OPS = set("+-*/%(),.;=<>[]{}&|^~?:")


def tokenize(sql: str) -> List[Token]:
    toks: List[Token] = []
    i, n = 0, len(sql)
    while i < n:
        c = sql[i]
        if c in " \t\r\n":
            i += 1
            continue
        if c == "-" and i + 1 < n and sql[i + 1] == "-":
            j = sql.find("\n", i)
            i = n if j < 0 else j + 1
            continue
        if c == "/" and i + 1 < n and sql[i + 1] == "*":
            j = sql.find("*/", i + 2)
            if j < 0:
                raise SqlError("unterminated block comment", sql, i)
            i = j + 2
            continue
        if c == "'" or c == '"':
            # Spark: single quotes = string literal; double quotes = string
            # literal too (unless ANSI mode, where they quote identifiers).
            # We treat double-quoted as string literals like Spark defaults.
            s, i = _read_quoted(sql, i, c)
            toks.append(Token("string", s, i))
            continue
        if c == "`":
            s, i = _read_quoted(sql, i, "`")
            toks.append(Token("ident", s, i))
            continue
        if c.isdigit() or (c == "." and i + 1 < n and sql[i + 1].isdigit()):
            j = i
            seen_dot = False
            seen_exp = False
            while j < n:
                ch = sql[j]
                if ch.isdigit():
                    j += 1
                elif ch == "." and not seen_dot and not seen_exp:
                    seen_dot = True
                    j += 1
                elif ch in "eE" and not seen_exp and j + 1 < n and (sql[j + 1].isdigit() or sql[j + 1] in "+-"):
                    seen_exp = True
                    j += 2 if sql[j + 1] in "+-" else 1
                else:
                    break
            # type suffix: 1L, 1.5D, 2S, 3Y, 1.0BD
            suffix = ""
            if j < n and sql[j : j + 2].upper() == "BD":
                suffix = "BD"
                j += 2
            elif j < n and sql[j].upper() in "LDSYF" and not (j + 1 < n and (sql[j + 1].isalnum() or sql[j + 1] == "_")):
                suffix = sql[j].upper()
                j += 1
            toks.append(Token("number", sql[i:j - len(suffix)] + ("#" + suffix if suffix else ""), i))
            i = j
            continue
        if c.isalpha() or c == "_":
            j = i
            while j < n and (sql[j].isalnum() or sql[j] == "_"):
                j += 1
            toks.append(Token("ident", sql[i:j], i))
            i = j
            continue
        matched = False
        for op in _MULTI_OPS:
            if sql.startswith(op, i):
                toks.append(Token("op", op, i))
                i += len(op)
                matched = True
                break
        if matched:
            continue
        if c in _SINGLE_OPS:
            toks.append(Token("op", c, i))
            i += 1
            continue
        if c == "!":
            toks.append(Token("op", "!", i))
            i += 1
            continue
        raise SqlError(f"unexpected character {c!r}", sql, i)
    toks.append(Token("eof", "", n))
    return toks


def _read_quoted(sql: str, i: int, q: str):
    j = i + 1
    out = []
    n = len(sql)
    while j < n:
        c = sql[j]
        if c == "\\" and q != "`" and j + 1 < n:
            esc = sql[j + 1]
            out.append({"n": "\n", "t": "\t", "r": "\r", "0": "\0"}.get(esc, esc))
            j += 2
            continue
        if c == q:
            if j + 1 < n and sql[j + 1] == q:  # doubled quote escape
                out.append(q)
                j += 2
                continue
            return "".join(out), j + 1
        out.append(c)
        j += 1
    raise SqlError("unterminated string literal", sql, i)
