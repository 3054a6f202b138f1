"""Regex -> NFA -> DFA compiler for the GPU multi-pattern scan kernel.

The reference's hot path is PCRE-class regex scanning per message
(redaction `registry.ts:31-150`, claims `claim-detector.ts`, entities
`patterns.ts`, cortex signals). On MI355X we compile each fixed pattern
set ONCE at startup into a single unanchored multi-pattern DFA over the
byte alphabet (reduced to equivalence classes), and scan message batches
with one GPU thread per message walking the table
(`csrc/pattern_scan.hip`). This module is the build-time compiler plus a
CPU reference scanner the kernel is tested against.

Supported subset (everything the builtin pattern families use):
  literals, '.', [classes] incl. ranges/negation, escapes \\d \\w \\s \\b
  and literal escapes, (?:...) / (...) groups, alternation, quantifiers
  * + ? {m} {m,} {m,n}, case-insensitive compile flag, and single-char
  lookbehind/lookahead `(?<!X)` / `(?!X)` which are converted into
  boundary guard characters (same hit semantics, unanchored scan).

Match semantics: "pattern i matches somewhere in the message" -> bit i of
the per-message hit mask. Greedy/leftmost details do not matter for hit
detection, so the classic subset construction is sufficient.
"""

from __future__ import annotations

import numpy as np
from typing import Dict, FrozenSet, List, Optional, Sequence, Set, Tuple

ANY = frozenset(range(256))
DIGITS = frozenset(range(ord("0"), ord("9") + 1))
WORD = frozenset(
    list(range(ord("a"), ord("z") + 1))
    + list(range(ord("A"), ord("Z") + 1))
    + list(range(ord("0"), ord("9") + 1))
    + [ord("_")]
)
SPACE = frozenset(ord(c) for c in " \t\n\r\f\v")
DOT = frozenset(b for b in range(256) if b != ord("\n"))

EPS = None  # epsilon edge label

# Bump on any change to the NFA/DFA construction so on-disk compiled-family
# caches (ops/pattern_sets.py:_family_hash) are invalidated; round-1's
# alternation pending-lookahead bug shipped through a stale cache.
DFA_COMPILER_VERSION = 4


class RegexError(ValueError):
    pass


# --------------------------------------------------------------------------
# Parser: regex subset -> AST
# node forms: ("char", frozenset), ("cat", [nodes]), ("alt", [nodes]),
#             ("star", node), ("plus", node), ("opt", node),
#             ("rep", node, m, n|None), ("bound",) word boundary,
#             ("nlb", frozenset) negative lookbehind, ("nla", frozenset)
# --------------------------------------------------------------------------


WORD_U8 = WORD | frozenset(range(128, 256))  # unicode_word: any UTF-8
# continuation/lead byte counts as a word byte (approximates re.UNICODE \w
# for hit detection over UTF-8 text)


class _Parser:
    def __init__(self, pattern: str, ignore_case: bool = False, unicode_word: bool = False):
        self.p = pattern
        self.i = 0
        self.ignore_case = ignore_case
        self.word = WORD_U8 if unicode_word else WORD

    def error(self, msg: str):
        raise RegexError(f"{msg} at {self.i} in {self.p!r}")

    def peek(self) -> Optional[str]:
        return self.p[self.i] if self.i < len(self.p) else None

    def next(self) -> str:
        c = self.p[self.i]
        self.i += 1
        return c

    def parse(self):
        node = self.parse_alt()
        if self.i != len(self.p):
            self.error("unexpected trailing input")
        return node

    def parse_alt(self):
        branches = [self.parse_cat()]
        while self.peek() == "|":
            self.next()
            branches.append(self.parse_cat())
        if len(branches) == 1:
            return branches[0]
        return ("alt", branches)

    def parse_cat(self):
        items = []
        while True:
            c = self.peek()
            if c is None or c in "|)":
                break
            items.append(self.parse_quant())
        if len(items) == 1:
            return items[0]
        return ("cat", items)

    def parse_quant(self):
        atom = self.parse_atom()
        while True:
            c = self.peek()
            if c == "*":
                self.next()
                atom = ("star", atom)
            elif c == "+":
                self.next()
                atom = ("plus", atom)
            elif c == "?":
                self.next()
                atom = ("opt", atom)
            elif c == "{":
                save = self.i
                self.next()
                digits = ""
                while self.peek() and self.peek().isdigit():
                    digits += self.next()
                if not digits:
                    self.i = save
                    break
                m = int(digits)
                n: Optional[int] = m
                if self.peek() == ",":
                    self.next()
                    digits2 = ""
                    while self.peek() and self.peek().isdigit():
                        digits2 += self.next()
                    n = int(digits2) if digits2 else None
                if self.peek() != "}":
                    self.i = save
                    break
                self.next()
                atom = ("rep", atom, m, n)
            else:
                break
            # non-greedy marker is irrelevant for hit detection
            if self.peek() == "?":
                self.next()
        return atom

    def _charset(self, chars: FrozenSet[int]) -> FrozenSet[int]:
        if not self.ignore_case:
            return chars
        out = set(chars)
        for b in chars:
            if ord("a") <= b <= ord("z"):
                out.add(b - 32)
            elif ord("A") <= b <= ord("Z"):
                out.add(b + 32)
        return frozenset(out)

    def parse_atom(self):
        c = self.next()
        if c == "(":
            if self.peek() == "?":
                self.next()
                k = self.peek()
                if k == ":":
                    self.next()
                elif k == "<":
                    self.next()
                    if self.peek() == "!":
                        self.next()
                        inner = self.parse_alt()
                        if self.peek() != ")":
                            self.error("unclosed lookbehind")
                        self.next()
                        cs = _single_charset(inner)
                        if cs is None:
                            self.error("only single-char lookbehind supported")
                        return ("nlb", self._charset(cs))
                    self.error("unsupported group (?<")
                elif k == "!":
                    self.next()
                    inner = self.parse_alt()
                    if self.peek() != ")":
                        self.error("unclosed lookahead")
                    self.next()
                    cs = _single_charset(inner)
                    if cs is None:
                        self.error("only single-char lookahead supported")
                    return ("nla", self._charset(cs))
                elif k == "=":
                    self.error("positive lookahead unsupported")
                else:
                    self.error("unsupported group flag")
            node = self.parse_alt()
            if self.peek() != ")":
                self.error("unclosed group")
            self.next()
            return node
        if c == "[":
            ascii_bytes, hi_cps, negated = self.parse_class()
            if negated:
                if hi_cps:
                    self.error("negated class with non-ASCII members unsupported")
                # fold case on the member set BEFORE negating (re.IGNORECASE
                # [^a-z] excludes A-Z as well)
                return ("char", frozenset(ANY - self._charset(frozenset(ascii_bytes))))
            branches = []
            if ascii_bytes:
                branches.append(("char", self._charset(frozenset(ascii_bytes))))
            for cp in sorted(hi_cps):
                branches.append(self._literal(chr(cp)))
            if not branches:
                self.error("empty class")
            return branches[0] if len(branches) == 1 else ("alt", branches)
        if c == ".":
            return ("char", DOT)
        if c == "\\":
            return self.parse_escape()
        if c in "^$":
            # ^ = start-of-input (re.search without MULTILINE); $ = end of
            # input, expressed as a pending (?![\s\S]) lookahead that only
            # matures at EOF
            return ("bof",) if c == "^" else ("nla", ANY)
        if c in "*+?":
            self.error("dangling quantifier")
        return self._literal(c)

    def _literal(self, c: str):
        """A literal character; non-ASCII expands to its UTF-8 byte
        sequence (with simple-case-fold variants when ignore_case)."""
        if ord(c) < 128:
            return ("char", self._charset(frozenset([ord(c)])))
        variants = {c}
        if self.ignore_case:
            # single-char case pairs only ('ß'.upper() == 'SS' is a
            # multi-char expansion re.IGNORECASE does not perform)
            variants |= {v for v in (c.lower(), c.upper()) if len(v) == 1}
        branches = []
        for v in sorted(variants):
            seq = v.encode("utf-8")
            if len(seq) == 1:
                branches.append(("char", frozenset(seq)))
            else:
                branches.append(("cat", [("char", frozenset([b])) for b in seq]))
        return branches[0] if len(branches) == 1 else ("alt", branches)

    def parse_escape(self):
        c = self.next()
        table = {
            "d": DIGITS,
            "D": ANY - DIGITS,
            "w": self.word,
            "W": ANY - self.word,
            "s": SPACE,
            "S": ANY - SPACE,
            "n": frozenset([10]),
            "t": frozenset([9]),
            "r": frozenset([13]),
        }
        if c in table:
            return ("char", table[c])
        if c == "b":
            return ("bound", self.word)
        return self._literal(c)

    def parse_class(self) -> Tuple[Set[int], Set[int], bool]:
        """Returns (ascii/escape byte set, non-ASCII literal codepoints,
        negated). Non-ASCII literal members become UTF-8 alternations in
        parse_atom; escape tables contribute BYTE values directly."""
        negate = False
        if self.peek() == "^":
            self.next()
            negate = True
        chars: Set[int] = set()
        hi_cps: Set[int] = set()
        first = True
        while True:
            c = self.peek()
            if c is None:
                self.error("unclosed class")
            if c == "]" and not first:
                self.next()
                break
            first = False
            self.next()
            if c == "\\":
                e = self.next()
                table = {"d": DIGITS, "w": self.word, "s": SPACE, "n": {10}, "t": {9}, "r": {13}}
                if e in table:
                    chars.update(table[e])
                    continue
                c = e
            lo = ord(c)
            if self.peek() == "-" and self.i + 1 < len(self.p) and self.p[self.i + 1] != "]":
                self.next()
                hi = ord(self.next())
                if hi < lo:
                    self.error("reversed class range")
                if hi - lo > 4096:
                    self.error("class range too wide")
                for cp in range(lo, hi + 1):
                    (chars if cp < 128 else hi_cps).add(cp)
            else:
                (chars if lo < 128 else hi_cps).add(lo)
        return chars, hi_cps, negate


def _single_charset(node) -> Optional[FrozenSet[int]]:
    if node[0] == "char":
        return node[1]
    if node[0] == "alt":
        out: Set[int] = set()
        for b in node[1]:
            cs = _single_charset(b)
            if cs is None:
                return None
            out.update(cs)
        return frozenset(out)
    return None


# --------------------------------------------------------------------------
# NFA (Thompson construction with word-boundary/lookaround guard edges)
# Transitions: list per state of (label, target); label is a frozenset of
# bytes, or ("eps",), or ("guard", kind, charset) where the guard tests the
# PREVIOUS byte (-1 at start): kind in {"prev_not_in", "prev_in"}.
# Lookahead (?!X) is rewritten as a guard on the NEXT byte evaluated lazily:
# we add an accept-variant reached only when the next byte is not in X or
# at end -> handled by deferring accept one step (see _compile_nla).
# --------------------------------------------------------------------------


class NFA:
    def __init__(self):
        self.trans: List[List[Tuple[object, int]]] = []
        # accepts: state -> set of (pattern_id, nla_charset|None)
        self.accepts: Dict[int, Set[Tuple[int, Optional[FrozenSet[int]]]]] = {}

    def new_state(self) -> int:
        self.trans.append([])
        return len(self.trans) - 1

    def add(self, src: int, label, dst: int) -> None:
        self.trans[src].append((label, dst))


Frontier = List[Tuple[int, Optional[FrozenSet[int]]]]


class _Unsupported(Exception):
    pass


def _restrict_first(node, allowed: FrozenSet[int]):
    """Restrict the FIRST consumed byte of `node` to `allowed`.

    Returns the rewritten node, or None when the restriction empties the
    branch. Raises _Unsupported when the first byte cannot be isolated
    (leading star/opt/guards) — callers fall back to the approximate
    prev-byte-only \b guard for those."""
    kind = node[0]
    if kind == "char":
        cs = node[1] & allowed
        return ("char", cs) if cs else None
    if kind == "cat":
        items = list(node[1])
        if not items:
            raise _Unsupported
        r = _restrict_first(items[0], allowed)
        if r is None:
            return None
        return ("cat", [r] + items[1:])
    if kind == "alt":
        branches = []
        for b in node[1]:
            r = _restrict_first(b, allowed)
            if r is not None:
                branches.append(r)
        if not branches:
            return None
        return branches[0] if len(branches) == 1 else ("alt", branches)
    if kind == "plus":
        r = _restrict_first(node[1], allowed)
        if r is None:
            return None
        return ("cat", [r, ("star", node[1])])
    if kind == "rep" and node[2] >= 1:
        _, inner, m, n = node
        r = _restrict_first(inner, allowed)
        if r is None:
            return None
        return ("cat", [r, ("rep", inner, m - 1, None if n is None else n - 1)])
    raise _Unsupported


def _rewrite_exact_bounds(items: list) -> list:
    """Rewrite mid-sequence \b into its exact two-sided form:
    (prev not word AND next word) OR (prev word AND next non-word),
    by intersecting the continuation's first-byte class. Fixes e.g.
    '\b[\w.%+-]+@...' wrongly hitting '.@example.com' (leading '.' is
    non-word, so re requires a word char before it — the old prev-only
    guard passed at BOF)."""
    out = []
    i = 0
    while i < len(items):
        it = items[i]
        if it[0] == "bound" and i + 1 < len(items):
            ws = it[1]
            nxt = items[i + 1]
            try:
                b_word = _restrict_first(nxt, ws)
                b_nonw = _restrict_first(nxt, ANY - ws)
            except _Unsupported:
                out.append(it)
                i += 1
                continue
            branches = []
            if b_word is not None:
                branches.append(("cat", [("nlb", ws), b_word]))
            if b_nonw is not None:
                branches.append(("cat", [("plb", ws), b_nonw]))
            if not branches:
                out.append(it)
                i += 1
                continue
            out.append(branches[0] if len(branches) == 1 else ("alt", branches))
            i += 2
            continue
        out.append(it)
        i += 1
    return out


def _join(nfa: NFA, frontier: Frontier) -> int:
    """Collapse an all-pendingless frontier into one state (error if any
    end still carries a pending negative lookahead — lookahead must be the
    last element of its branch)."""
    if any(p is not None for _e, p in frontier):
        raise RegexError("lookahead must be last in a branch")
    if len(frontier) == 1:
        return frontier[0][0]
    end = nfa.new_state()
    for e, _p in frontier:
        nfa.add(e, ("eps",), end)
    return end


def _build(nfa: NFA, node, start: int) -> Frontier:
    """Build node starting at `start`; return the accept frontier — a list
    of (end_state, pending_nla) pairs. A trailing (?!X) becomes a pending
    negative-lookahead attached to that branch's accept ONLY: alternation
    branches keep separate ends when their pendings differ, so e.g.
    `\\bDAN\\b|do anything now` does not leak branch 1's trailing \\b onto
    branch 2 (round-1 advisor finding, dfa alternation bug)."""
    kind = node[0]
    if kind == "char":
        end = nfa.new_state()
        nfa.add(start, node[1], end)
        return [(end, None)]
    if kind == "cat":
        items = list(node[1])
        # trailing \b means "next byte is not word (or EOF)" -> lookahead
        if items and items[-1][0] == "bound":
            items[-1] = ("nla", items[-1][1])
        items = _rewrite_exact_bounds(items)
        cur = start
        frontier: Frontier = [(start, None)]
        for item in items:
            cur = _join(nfa, frontier)
            frontier = _build(nfa, item, cur)
        return frontier
    if kind == "alt":
        out: Frontier = []
        plain: Frontier = []
        for b in node[1]:
            s = nfa.new_state()
            nfa.add(start, ("eps",), s)
            for e, pend in _build(nfa, b, s):
                (plain if pend is None else out).append((e, pend))
        if plain:
            # merge pendingless branch ends into one shared state (keeps the
            # subset construction small); pending ends stay distinct.
            out.append((_join(nfa, plain), None))
        return out
    if kind in ("star", "plus", "opt"):
        inner = node[1]
        s = nfa.new_state()
        e = nfa.new_state()
        nfa.add(start, ("eps",), s)
        ie = _join(nfa, _build(nfa, inner, s))
        nfa.add(ie, ("eps",), e)
        if kind in ("star", "plus"):
            nfa.add(ie, ("eps",), s)
        if kind in ("star", "opt"):
            nfa.add(start, ("eps",), e)
        return [(e, None)]
    if kind == "rep":
        _, inner, m, n = node
        cur = start
        for _i in range(m):
            cur = _join(nfa, _build(nfa, inner, cur))
        if n is None:
            # {m,} -> m copies + star
            cur = _join(nfa, _build(nfa, ("star", inner), cur))
        else:
            for _i in range(n - m):
                cur = _join(nfa, _build(nfa, ("opt", inner), cur))
        return [(cur, None)]
    if kind == "bound":
        # \b before a word char: previous byte must NOT be word (or start).
        # \b after a word char: next byte must not be word — approximated by
        # the prev-guard form on the following edge; for trailing \b we
        # attach a pending lookahead on the word set (node[1]: WORD, or
        # WORD_U8 under unicode_word).
        end = nfa.new_state()
        nfa.add(start, ("guard", "boundary", node[1]), end)
        return [(end, None)]
    if kind == "bof":
        # start-of-input: passes only when there is no previous byte
        end = nfa.new_state()
        nfa.add(start, ("guard", "bof", frozenset()), end)
        return [(end, None)]
    if kind == "nlb":
        end = nfa.new_state()
        nfa.add(start, ("guard", "prev_not_in", node[1]), end)
        return [(end, None)]
    if kind == "plb":
        # positive lookbehind guard: previous byte must be in the set
        end = nfa.new_state()
        nfa.add(start, ("guard", "prev_in", node[1]), end)
        return [(end, None)]
    if kind == "nla":
        # pending negative lookahead — resolved at accept time
        return [(start, node[1])]
    raise RegexError(f"unknown node {kind}")


# --------------------------------------------------------------------------
# Subset construction over an extended state: because guards depend on the
# PREVIOUS byte, we make the DFA alphabet the byte being consumed and keep
# "which NFA states are live" as usual, but epsilon/guard closure is
# computed per (stateset, prev_byte_class). To keep this finite we resolve
# guards at transition time: when consuming byte b from DFA state S, we
# first take guard edges valid given prev byte p — p IS the byte consumed
# on the PREVIOUS step, which the DFA remembers via its state identity:
# we tag each DFA state with the equivalence class of the byte that led
# into it (start tag = BOF). This multiplies states by at most the number
# of distinct guard-relevant classes (word / digit membership: 4 tags).
# --------------------------------------------------------------------------


class DFA:
    """Tables ready for the GPU kernel:
    - next_state: uint16/uint32 [n_states, n_classes]
    - accept_mask: uint64 [n_states] (bit i = pattern i matched)
    - byte_class: uint8 [256]
    """

    def __init__(self, next_state: np.ndarray, accept_mask: np.ndarray, byte_class: np.ndarray):
        self.next_state = next_state
        self.accept_mask = accept_mask
        self.byte_class = byte_class

    @property
    def n_states(self) -> int:
        return self.next_state.shape[0]

    @property
    def n_classes(self) -> int:
        return self.next_state.shape[1]

    def scan(self, data: bytes) -> int:
        """CPU reference scanner: returns the hit bitmask. Matches the GPU
        kernel in csrc/pattern_scan.hip step for step."""
        state = 0
        mask = 0
        nxt = self.next_state
        acc = self.accept_mask
        bc = self.byte_class
        mask |= int(acc[state])
        for b in data:
            state = int(nxt[state, bc[b]])
            mask |= int(acc[state])
        return mask

    def nbytes(self) -> int:
        return self.next_state.nbytes + self.accept_mask.nbytes + self.byte_class.nbytes


def _closure(nfa: NFA, states: FrozenSet[int], prev_is: Dict[str, bool]) -> FrozenSet[int]:
    """Epsilon + guard closure given predicates about the previous byte."""
    out = set(states)
    stack = list(states)
    while stack:
        s = stack.pop()
        for label, dst in nfa.trans[s]:
            if isinstance(label, tuple):
                ok = False
                if label[0] == "eps":
                    ok = True
                elif label[0] == "guard":
                    _, kind, cs = label
                    if kind == "prev_not_in":
                        ok = not prev_is["in:" + _cs_key(cs)]
                    elif kind == "prev_in":
                        ok = prev_is["in:" + _cs_key(cs)]
                    elif kind == "boundary":
                        # \b before consuming a word char: prev not word
                        ok = not prev_is["in:" + _cs_key(cs)]
                    elif kind == "bof":
                        ok = prev_is.get("bof", False)
                if ok and dst not in out:
                    out.add(dst)
                    stack.append(dst)
    return frozenset(out)


_cs_keys: Dict[FrozenSet[int], str] = {}


def _cs_key(cs: FrozenSet[int]) -> str:
    key = _cs_keys.get(cs)
    if key is None:
        key = str(len(_cs_keys))
        _cs_keys[cs] = key
    return key


def compile_patterns(
    patterns: Sequence[Tuple[str, int]],
    ignore_case_ids: Optional[Set[int]] = None,
    max_states: int = 60000,
    unicode_word: bool = False,
) -> DFA:
    """Compile [(regex, pattern_id)] into one unanchored multi-pattern DFA.

    pattern_id is the bit set in the accept mask (0..63); several patterns
    may share a bit (the cortex family ORs 10 languages into one signal
    bit). unicode_word widens \w/\b to treat any byte >= 0x80 as a word
    byte (UTF-8 text under re.UNICODE semantics, hit-detection grade).
    """
    ignore_case_ids = ignore_case_ids or set()
    nfa = NFA()
    root = nfa.new_state()

    guard_sets: Set[FrozenSet[int]] = {WORD_U8 if unicode_word else WORD}
    for pattern, pid in patterns:
        if pid < 0 or pid > 63:
            raise RegexError("pattern_id must fit a u64 bitmask")
        ast = _Parser(pattern, ignore_case=pid in ignore_case_ids,
                      unicode_word=unicode_word).parse()
        s = nfa.new_state()
        nfa.add(root, ("eps",), s)
        for end, pending_nla in _build(nfa, ast, s):
            nfa.accepts.setdefault(end, set()).add((pid, pending_nla))
            if pending_nla is not None:
                guard_sets.add(pending_nla)
    # collect guard charsets used
    for edges in nfa.trans:
        for label, _ in edges:
            if isinstance(label, tuple) and label[0] == "guard":
                guard_sets.add(label[2])

    # byte equivalence classes: two bytes are equivalent if no transition
    # charset or guard set distinguishes them
    sig_sets: List[FrozenSet[int]] = []
    for edges in nfa.trans:
        for label, _ in edges:
            if isinstance(label, frozenset):
                sig_sets.append(label)
    sig_sets.extend(guard_sets)
    sigs: Dict[int, Tuple[bool, ...]] = {}
    for b in range(256):
        sigs[b] = tuple(b in s for s in sig_sets)
    classes: Dict[Tuple[bool, ...], int] = {}
    byte_class = np.zeros(256, dtype=np.uint8)
    for b in range(256):
        cid = classes.setdefault(sigs[b], len(classes))
        byte_class[b] = cid
    n_classes = len(classes)
    class_rep = [0] * n_classes
    for b in range(255, -1, -1):
        class_rep[byte_class[b]] = b

    # guard predicate context per incoming byte class (+ BOF)
    _ctx_cache: Dict[Optional[int], Dict[str, bool]] = {}

    def prev_ctx(byte: Optional[int]) -> Dict[str, bool]:
        if byte in _ctx_cache:
            return _ctx_cache[byte]
        ctx = {"bof": byte is None}
        for cs in guard_sets:
            ctx["in:" + _cs_key(cs)] = byte is not None and byte in cs
        _ctx_cache[byte] = ctx
        return ctx

    def accept_mask_of(states: FrozenSet[int], next_byte_cls: Optional[int]) -> int:
        """Accepts valid in this state-set. Pending negative lookaheads are
        resolved against the NEXT byte — since the DFA reports hits as it
        consumes, we defer those accepts: an accept with (?!X) fires in the
        successor state when the consumed byte is not in X, or at EOF. To
        keep the table simple we fold this in during construction: see
        deferred handling below."""
        mask = 0
        for s in states:
            for pid, nla in nfa.accepts.get(s, ()):
                if nla is None:
                    mask |= 1 << pid
        return mask

    # DFA construction: state = (frozenset of NFA states, guard profile of
    # the incoming byte or BOF, fired-deferred-accept mask). The fired mask
    # carries accepts whose trailing (?!X) matured on the incoming
    # transition — it must be part of state identity, NOT painted onto a
    # shared successor state.
    start_key = (_closure(nfa, frozenset([root]), prev_ctx(None)), -1, 0)
    index: Dict[Tuple[FrozenSet[int], int, int], int] = {start_key: 0}
    order: List[Tuple[FrozenSet[int], int, int]] = [start_key]
    next_rows: List[List[int]] = []
    accept_rows: List[int] = []
    # deferred (?!X) accepts: state -> set of (pid, X)
    deferred: List[Set[Tuple[int, FrozenSet[int]]]] = []

    # Precompute per-NFA-state char edges as (class-id tuple, dst) so the
    # hot loop touches each edge once per DFA state, not once per class.
    edge_classes: List[List[Tuple[Tuple[int, ...], int]]] = []
    _label_classes: Dict[FrozenSet[int], Tuple[int, ...]] = {}
    for s in range(len(nfa.trans)):
        lst = []
        for label, dst in nfa.trans[s]:
            if isinstance(label, frozenset):
                cls_tuple = _label_classes.get(label)
                if cls_tuple is None:
                    cls_tuple = tuple(sorted({int(byte_class[b]) for b in label}))
                    _label_classes[label] = cls_tuple
                lst.append((cls_tuple, dst))
        edge_classes.append(lst)

    # --- counter-chain domination -------------------------------------
    # A pattern tail like PREFIX [cls]{80} expands to a uniform chain of
    # NFA states ending in an accept with no outgoing edges. Unanchored
    # scanning keeps spawning younger counter instances at every PREFIX
    # occurrence; tracking every active position explodes the subset
    # construction (C(80, k) state sets). For HIT detection the oldest
    # instance dominates: all instances see the same bytes, die together
    # on a charset break, and the oldest accepts first. So inside one
    # uniform chain we keep only the most-advanced state. Sound only when
    # the chain end is accepting with no outgoing edges (suffix-free) —
    # patterns with a live suffix (e.g. phone's \d{6,14}(?!\d)) are not
    # chains by this definition and keep full tracking.
    chain_id: Dict[int, int] = {}
    chain_pos: Dict[int, int] = {}
    n_states_nfa = len(nfa.trans)
    is_accept = set(nfa.accepts.keys())
    # uniform link: s -> t, single outgoing char edge, s not accepting
    link: Dict[int, Tuple[FrozenSet[int], int]] = {}
    incoming_links: Dict[int, int] = {}
    for s in range(n_states_nfa):
        edges = nfa.trans[s]
        if s not in is_accept and len(edges) == 1 and isinstance(edges[0][0], frozenset):
            link[s] = (edges[0][0], edges[0][1])
            incoming_links[edges[0][1]] = incoming_links.get(edges[0][1], 0) + 1
    next_chain = 0
    for s in list(link.keys()):
        if s in chain_id:
            continue
        # find chain head: walk back not possible cheaply; start anywhere and
        # only accept chains ending at an accept state with no outgoing edges
        seq = [s]
        cs = link[s][0]
        cur = link[s][1]
        ok = True
        seen_local = {s}
        while cur in link:
            if link[cur][0] != cs or cur in seen_local:
                ok = False
                break
            seq.append(cur)
            seen_local.add(cur)
            cur = link[cur][1]
        if not ok:
            continue
        # cur is the chain end: must be accepting with no outgoing edges and
        # no pending lookahead on its accepts
        if cur not in is_accept or nfa.trans[cur] or any(
            nla is not None for _pid, nla in nfa.accepts.get(cur, ())
        ):
            continue
        if len(seq) < 4:
            continue  # not worth canonicalizing
        cid = next_chain
        next_chain += 1
        for pos, st in enumerate(seq):
            chain_id[st] = cid
            chain_pos[st] = pos
        chain_id[cur] = cid
        chain_pos[cur] = len(seq)

    def canonicalize(states: FrozenSet[int]) -> FrozenSet[int]:
        if not chain_id:
            return states
        best: Dict[int, int] = {}
        rest: List[int] = []
        for st in states:
            cid = chain_id.get(st)
            if cid is None:
                rest.append(st)
            else:
                prev = best.get(cid)
                if prev is None or chain_pos[st] > chain_pos[prev]:
                    best[cid] = st
        return frozenset(rest + list(best.values()))

    # Guards only observe membership of the previous byte in guard_sets, so
    # DFA states need only be split by that small "guard profile", not by
    # the full incoming byte class (which would multiply states ~40x).
    guard_list = sorted(guard_sets, key=_cs_key)
    profile_of_cls: List[int] = []
    for cls in range(n_classes):
        b = class_rep[cls]
        prof = 0
        for gi, cs in enumerate(guard_list):
            if b in cs:
                prof |= 1 << gi
        profile_of_cls.append(prof)
    BOF_PROFILE = -1

    _closure_cache: Dict[Tuple[FrozenSet[int], int], FrozenSet[int]] = {}

    def closure_cached(states: FrozenSet[int], cls: int) -> Tuple[FrozenSet[int], int]:
        prof = profile_of_cls[cls]
        key = (states, prof)
        out = _closure_cache.get(key)
        if out is None:
            out = canonicalize(_closure(nfa, states, prev_ctx(class_rep[cls])))
            _closure_cache[key] = out
        return out, prof

    i = 0
    while i < len(order):
        states, in_cls, fired = order[i]
        i += 1
        row = [0] * n_classes
        acc = accept_mask_of(states, None) | fired
        defs: Set[Tuple[int, FrozenSet[int]]] = set()
        for s in states:
            for pid, nla in nfa.accepts.get(s, ()):
                if nla is not None:
                    defs.add((pid, nla))
        moved_per_cls: List[Set[int]] = [set() for _ in range(n_classes)]
        for s in states:
            for cls_tuple, dst in edge_classes[s]:
                for cls in cls_tuple:
                    moved_per_cls[cls].add(dst)
        for cls in range(n_classes):
            moved = moved_per_cls[cls]
            moved.add(root)  # unanchored scan: root stays alive
            closed, prof = closure_cached(frozenset(moved), cls)
            b = class_rep[cls]
            fired_next = 0
            for pid, X in defs:
                if b not in X:
                    fired_next |= 1 << pid
            key = (closed, prof, fired_next)
            j = index.get(key)
            if j is None:
                j = len(order)
                if j >= max_states:
                    raise RegexError(f"DFA exceeds {max_states} states")
                index[key] = j
                order.append(key)
            row[cls] = j
        next_rows.append(row)
        accept_rows.append(acc)
        deferred.append(defs)

    n = len(order)
    next_state = np.array(next_rows, dtype=np.uint32 if n > 65535 else np.uint16)
    accept = np.array(accept_rows, dtype=np.uint64)

    # deferred (?!X) at EOF: pending accepts mature at end of input
    eof_mask = np.zeros(n, dtype=np.uint64)
    for s in range(n):
        for pid, X in deferred[s]:
            eof_mask[s] |= np.uint64(1 << pid)

    dfa = DFA(next_state, accept, byte_class)
    dfa.eof_mask = eof_mask  # type: ignore[attr-defined]
    return dfa


class MultiDFA:
    """A family compiled as several small DFAs (counter products across
    patterns otherwise explode the subset construction). The GPU kernel
    walks each sub-DFA over the message; hit masks OR together.
    Tables are concatenated for upload: see `pack()`."""

    def __init__(self, dfas: List[DFA]):
        self.dfas = dfas

    @property
    def n_states(self) -> int:
        return sum(d.n_states for d in self.dfas)

    def nbytes(self) -> int:
        return sum(d.nbytes() for d in self.dfas)

    def scan(self, data: bytes) -> int:
        mask = 0
        for d in self.dfas:
            mask |= scan_with_eof(d, data)
        return mask

    def pack(self):
        """Concatenate tables into flat arrays for the GPU kernel
        (csrc/pattern_scan.hip):
          next:       uint16 flat — each sub-DFA's [n_states, n_classes]
                      class-transition table (LOCAL state ids)
          accept,eof: uint64 [sum_states] (indexed by state_base + state)
          class_maps: uint8 [n_dfas, 256] byte -> class
          meta:       int32 [n_dfas, 4] = (next_base_in_u16s, state_base,
                      n_classes, class_map_row)
        The kernel keeps class_maps in LDS and walks `next` out of L2
        (the tables are a few hundred KB — L2-resident)."""
        total = self.n_states
        accept = np.zeros(total, dtype=np.uint64)
        eof = np.zeros(total, dtype=np.uint64)
        meta = np.zeros((len(self.dfas), 4), dtype=np.int32)
        class_maps = np.zeros((len(self.dfas), 256), dtype=np.uint8)
        next_parts = []
        state_base = 0
        next_base = 0
        for i, d in enumerate(self.dfas):
            if d.n_states > 65535:
                raise RegexError("sub-DFA exceeds uint16 state ids")
            meta[i] = (next_base, state_base, d.n_classes, i)
            class_maps[i] = d.byte_class
            next_parts.append(d.next_state.astype(np.uint16).reshape(-1))
            accept[state_base : state_base + d.n_states] = d.accept_mask
            e = getattr(d, "eof_mask", None)
            if e is not None:
                eof[state_base : state_base + d.n_states] = e
            state_base += d.n_states
            next_base += d.n_states * d.n_classes
        return {
            "next": np.concatenate(next_parts),
            "accept": accept,
            "eof": eof,
            "class_maps": class_maps,
            "meta": meta,
        }


def compile_multi(
    patterns: Sequence[Tuple[str, int]],
    ignore_case_ids: Optional[Set[int]] = None,
    per_dfa_state_budget: int = 3000,
    unicode_word: bool = False,
) -> MultiDFA:
    """Greedily pack patterns into as few DFAs as fit the state budget."""
    ignore_case_ids = ignore_case_ids or set()
    dfas: List[DFA] = []
    group: List[Tuple[str, int]] = []
    group_dfa: Optional[DFA] = None

    for pat in patterns:
        trial = group + [pat]
        try:
            d = compile_patterns(trial, ignore_case_ids=ignore_case_ids,
                                 max_states=per_dfa_state_budget, unicode_word=unicode_word)
        except RegexError:
            d = None
        if d is None:
            if group_dfa is not None:
                dfas.append(group_dfa)
            group = [pat]
            group_dfa = compile_patterns(group, ignore_case_ids=ignore_case_ids,
                                         max_states=per_dfa_state_budget * 8,
                                         unicode_word=unicode_word)
        else:
            group = trial
            group_dfa = d
    if group_dfa is not None:
        dfas.append(group_dfa)
    return MultiDFA(dfas)


def scan_with_eof(dfa: DFA, data: bytes) -> int:
    """Reference scan including deferred-EOF accepts."""
    state = 0
    mask = int(dfa.accept_mask[0])
    for b in data:
        state = int(dfa.next_state[state, dfa.byte_class[b]])
        mask |= int(dfa.accept_mask[state])
    eof = getattr(dfa, "eof_mask", None)
    if eof is not None:
        mask |= int(eof[state])
    return mask
