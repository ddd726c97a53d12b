"""Cypher lexer.

Single proper tokenizer feeding a recursive-descent parser — replaces the
reference's dual-parser setup (hand-rolled keyword/regex "Nornic" parser +
ANTLR validation parser, reference pkg/cypher/parser.go + antlr/) with one
AST pipeline.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List

KEYWORDS = {
    "MATCH", "OPTIONAL", "WHERE", "RETURN", "CREATE", "MERGE", "SET", "REMOVE",
    "DELETE", "DETACH", "WITH", "UNWIND", "AS", "ORDER", "BY", "SKIP", "LIMIT",
    "ASC", "ASCENDING", "DESC", "DESCENDING", "DISTINCT", "AND", "OR", "XOR",
    "NOT", "IN", "STARTS", "ENDS", "CONTAINS", "IS", "NULL", "TRUE", "FALSE",
    "CALL", "YIELD", "UNION", "ALL", "ON", "CASE", "WHEN", "THEN", "ELSE",
    "END", "EXISTS", "COUNT", "ANY", "NONE", "SINGLE", "EXPLAIN", "PROFILE",
    "FOREACH", "USING", "INDEX", "DROP", "CONSTRAINT", "UNIQUE", "ASSERT",
    "SHOW", "DATABASE", "DATABASES",
}


@dataclass
class Token:
    kind: str   # KW, IDENT, INT, FLOAT, STRING, OP, PARAM, EOF
    value: str
    pos: int
    raw: str = ""   # original source text (case-preserving, KW tokens)


class CypherSyntaxError(Exception):
    pass


def tokenize(text: str) -> List[Token]:
    toks: List[Token] = []
    i, n = 0, len(text)
    while i < n:
        c = text[i]
        if c.isspace():
            i += 1
            continue
        if text.startswith("//", i):
            j = text.find("\n", i)
            i = n if j < 0 else j + 1
            continue
        if text.startswith("/*", i):
            j = text.find("*/", i + 2)
            if j < 0:
                raise CypherSyntaxError("unterminated comment")
            i = j + 2
            continue
        if c == "$":  # parameter
            j = i + 1
            if j < n and text[j] == "{":
                k = text.find("}", j)
                if k < 0:
                    raise CypherSyntaxError("unterminated ${param}")
                toks.append(Token("PARAM", text[j + 1:k], i))
                i = k + 1
                continue
            while j < n and (text[j].isalnum() or text[j] == "_"):
                j += 1
            if j == i + 1:
                raise CypherSyntaxError(f"bad parameter at {i}")
            toks.append(Token("PARAM", text[i + 1:j], i))
            i = j
            continue
        if c in "\"'":
            j = i + 1
            buf = []
            while j < n:
                if text[j] == "\\" and j + 1 < n:
                    esc = text[j + 1]
                    buf.append({"n": "\n", "t": "\t", "r": "\r", "\\": "\\",
                                "'": "'", '"': '"', "b": "\b", "f": "\f",
                                "u": "\\u"}.get(esc, esc))
                    if esc == "u" and j + 5 < n:
                        buf[-1] = chr(int(text[j + 2:j + 6], 16))
                        j += 4
                    j += 2
                    continue
                if text[j] == c:
                    break
                buf.append(text[j])
                j += 1
            if j >= n:
                raise CypherSyntaxError("unterminated string")
            toks.append(Token("STRING", "".join(buf), i))
            i = j + 1
            continue
        if c == "`":  # escaped identifier
            j = text.find("`", i + 1)
            if j < 0:
                raise CypherSyntaxError("unterminated `identifier`")
            toks.append(Token("IDENT", text[i + 1:j], i))
            i = j + 1
            continue
        if c.isdigit() or (c == "." and i + 1 < n and text[i + 1].isdigit()):
            j = i
            isf = False
            while j < n and (text[j].isdigit() or text[j] in ".eExX+-abcdefABCDEF"):
                if text[j] in ".eE":
                    # stop at '..' range operator
                    if text[j] == "." and text.startswith("..", j):
                        break
                    if text[j] in "eE" and not (j + 1 < n and (text[j + 1].isdigit() or text[j + 1] in "+-")):
                        break
                    isf = isf or text[j] == "." or text[j] in "eE"
                elif text[j] in "+-" and text[j - 1] not in "eE":
                    break
                elif text[j] in "xX" and not text.startswith("0", i):
                    break
                elif text[j] in "abcdefABCDEF" and not text[i:i+2].lower() == "0x":
                    break
                j += 1
            lit = text[i:j]
            if lit.lower().startswith("0x"):
                toks.append(Token("INT", str(int(lit, 16)), i))
            elif isf:
                toks.append(Token("FLOAT", lit, i))
            else:
                toks.append(Token("INT", lit, i))
            i = j
            continue
        if c.isalpha() or c == "_":
            j = i
            while j < n and (text[j].isalnum() or text[j] == "_"):
                j += 1
            word = text[i:j]
            if word.upper() in KEYWORDS:
                t = Token("KW", word.upper(), i)
                t.raw = word  # original case for label/name positions
                toks.append(t)
            else:
                toks.append(Token("IDENT", word, i))
            i = j
            continue
        # operators
        for op in ("<>", "!=", "<=", ">=", "=~", "..", "+=", "->", "<-"):
            if text.startswith(op, i):
                toks.append(Token("OP", op, i))
                i += len(op)
                break
        else:
            if c in "()[]{}.,:;|=<>+-*/%^!":
                toks.append(Token("OP", c, i))
                i += 1
            else:
                raise CypherSyntaxError(f"unexpected character {c!r} at {i}")
    toks.append(Token("EOF", "", n))
    return toks
