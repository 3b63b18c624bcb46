"""SQL tokenizer shared by the SELECT parser and the FugueSQL dialect
parser.  New implementation (the reference delegates to ANTLR via the
external ``fugue-sql-antlr`` package)."""
import re
from typing import List, NamedTuple, Optional


class Token(NamedTuple):
    kind: str  # KW, NAME, NUMBER, STRING, OP, PUNCT
    value: str
    pos: int
    end: int = -1  # source end offset (pos + raw length; value may be
    #              shorter than the raw text for `quoted` names)

    @property
    def upper(self) -> str:
        return self.value.upper()

    @property
    def src_end(self) -> int:
        return self.end if self.end >= 0 else self.pos + len(self.value)


_TOKEN_RE = re.compile(
    r"""
    (?P<WS>\s+)
  | (?P<COMMENT>--[^\n]*|/\*.*?\*/)
  | (?P<STRING>'(?:[^']|'')*'|"(?:[^"]|"")*")
  | (?P<NUMBER>\d+\.\d*|\.\d+|\d+)
  | (?P<NAME>[A-Za-z_][A-Za-z_0-9]*|`[^`]+`)
  | (?P<OP><=|>=|<>|!=|==|=|<|>|\|\||[+\-*/%])
  | (?P<PUNCT>[(),.;:\[\]{}])
    """,
    re.VERBOSE | re.DOTALL,
)


def tokenize(sql: str) -> List[Token]:
    """Tokenize with a small memo: identical statement texts (the common
    case for repeated queries) skip the regex scan.  Callers treat the
    token list as read-only (TokenStream only advances an index)."""
    hit = _TOKEN_CACHE.get(sql)
    if hit is not None:
        return hit
    tokens = _tokenize_uncached(sql)
    if len(_TOKEN_CACHE) > 256:
        _TOKEN_CACHE.clear()
    _TOKEN_CACHE[sql] = tokens
    return tokens


_TOKEN_CACHE: dict = {}


def _tokenize_uncached(sql: str) -> List[Token]:
    tokens: List[Token] = []
    pos = 0
    while pos < len(sql):
        m = _TOKEN_RE.match(sql, pos)
        if m is None:
            raise SyntaxError(f"can't tokenize SQL at {pos}: {sql[pos:pos+30]!r}")
        kind = m.lastgroup or ""
        value = m.group()
        if kind not in ("WS", "COMMENT"):
            if kind == "NAME" and value.startswith("`"):
                value = value[1:-1]
            tokens.append(Token(kind, value, pos, m.end()))
        pos = m.end()
    return tokens


class TokenStream:
    def __init__(self, tokens: List[Token]):
        self.tokens = tokens
        self.pos = 0

    def peek(self, offset: int = 0) -> Optional[Token]:
        i = self.pos + offset
        return self.tokens[i] if i < len(self.tokens) else None

    def next(self) -> Token:
        t = self.peek()
        if t is None:
            raise SyntaxError("unexpected end of SQL")
        self.pos += 1
        return t

    def match_kw(self, *kws: str) -> bool:
        t = self.peek()
        return t is not None and t.kind == "NAME" and t.upper in kws

    def take_kw(self, *kws: str) -> bool:
        if self.match_kw(*kws):
            self.next()
            return True
        return False

    def expect_kw(self, kw: str) -> None:
        if not self.take_kw(kw):
            t = self.peek()
            raise SyntaxError(f"expected {kw}, got {t.value if t else 'EOF'}")

    def match_punct(self, p: str) -> bool:
        t = self.peek()
        return t is not None and t.kind in ("PUNCT", "OP") and t.value == p

    def take_punct(self, p: str) -> bool:
        if self.match_punct(p):
            self.next()
            return True
        return False

    def expect_punct(self, p: str) -> None:
        if not self.take_punct(p):
            t = self.peek()
            raise SyntaxError(f"expected {p!r}, got {t.value if t else 'EOF'}")

    @property
    def eof(self) -> bool:
        return self.pos >= len(self.tokens)
