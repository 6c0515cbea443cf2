"""Regex-constrained decoding: regex → NFA → lazy DFA → per-DFA-state
token masks via a vocab-trie × DFA product walk.

Role of the reference's outlines/xgrammar regex backends
(vllm/v1/structured_output/backend_outlines.py,
outlines_core's Index): the same compile-to-FSM + per-state vocab mask
design, built offline-capable in pure Python. Masks and (state, token)
advances are memoised, and the vocabulary is walked as a character trie
so shared token prefixes are processed once per DFA state.

Supported regex subset: literals, escapes (\\d \\w \\s \\D \\W \\S and
single-char escapes), ``.``, character classes ``[...]`` with ranges and
negation, groups ``( )`` / ``(?: )``, alternation ``|``, and the
quantifiers ``* + ? {m} {m,} {m,n}``.
"""

from __future__ import annotations

from typing import Optional

# --------------------------------------------------------------------------
# Character classes: sorted, merged (lo, hi) codepoint ranges.
# --------------------------------------------------------------------------

MAX_CP = 0x10FFFF


def _merge(ranges):
    out = []
    for lo, hi in sorted(ranges):
        if out and lo <= out[-1][1] + 1:
            out[-1] = (out[-1][0], max(out[-1][1], hi))
        else:
            out.append((lo, hi))
    return tuple(out)


def _negate(ranges):
    out = []
    prev = 0
    for lo, hi in ranges:
        if lo > prev:
            out.append((prev, lo - 1))
        prev = hi + 1
    if prev <= MAX_CP:
        out.append((prev, MAX_CP))
    return tuple(out)


_D = (((ord("0"), ord("9")),))
_W = _merge([(ord("a"), ord("z")), (ord("A"), ord("Z")),
             (ord("0"), ord("9")), (ord("_"), ord("_"))])
_S = _merge([(9, 13), (32, 32)])
_DOT = _negate(((10, 10),))  # any char but newline


class CharClass:
    __slots__ = ("ranges",)

    def __init__(self, ranges):
        self.ranges = _merge(ranges)

    def matches(self, cp: int) -> bool:
        # Binary search over few ranges; linear is fine (classes are small).
        for lo, hi in self.ranges:
            if lo <= cp <= hi:
                return True
            if cp < lo:
                return False
        return False


# --------------------------------------------------------------------------
# Regex parser → NFA (Thompson construction).
# --------------------------------------------------------------------------


class _NFA:
    def __init__(self):
        self.eps: list[list[int]] = []  # state -> eps targets
        self.trans: list[list[tuple[CharClass, int]]] = []  # state -> edges
        self.start = self._new()
        self.accept: int = -1

    def _new(self) -> int:
        self.eps.append([])
        self.trans.append([])
        return len(self.eps) - 1


_ESCAPES = {
    "d": _D, "w": _W, "s": _S,
    "D": _negate(_D), "W": _negate(_W), "S": _negate(_S),
}
_CTRL = {"n": "\n", "t": "\t", "r": "\r", "f": "\f", "v": "\v",
         "0": "\0", "a": "\a", "b": "\b"}


class _Parser:
    """Recursive-descent regex parser producing an NFA fragment tree."""

    def __init__(self, pattern: str):
        self.p = pattern
        self.i = 0

    def _peek(self) -> Optional[str]:
        return self.p[self.i] if self.i < len(self.p) else None

    def _next(self) -> str:
        ch = self.p[self.i]
        self.i += 1
        return ch

    # AST nodes: ("lit", CharClass) | ("cat", [..]) | ("alt", [..])
    #          | ("rep", node, min, max|None) | ("empty",)
    def parse(self):
        node = self._alt()
        if self.i != len(self.p):
            raise ValueError(f"regex parse error at {self.i}: {self.p!r}")
        return node

    def _alt(self):
        branches = [self._seq()]
        while self._peek() == "|":
            self._next()
            branches.append(self._seq())
        return branches[0] if len(branches) == 1 else ("alt", branches)

    def _seq(self):
        items = []
        while self._peek() not in (None, "|", ")"):
            items.append(self._repeat())
        if not items:
            return ("empty",)
        return items[0] if len(items) == 1 else ("cat", items)

    def _repeat(self):
        node = self._atom()
        while True:
            ch = self._peek()
            if ch == "*":
                self._next()
                node = ("rep", node, 0, None)
            elif ch == "+":
                self._next()
                node = ("rep", node, 1, None)
            elif ch == "?":
                self._next()
                node = ("rep", node, 0, 1)
            elif ch == "{":
                save = self.i
                self._next()
                body = ""
                while self._peek() not in (None, "}"):
                    body += self._next()
                if self._peek() != "}":
                    self.i = save
                    break
                self._next()
                try:
                    if "," in body:
                        lo_s, hi_s = body.split(",", 1)
                        lo = int(lo_s)
                        hi = int(hi_s) if hi_s.strip() else None
                    else:
                        lo = hi = int(body)
                except ValueError:
                    self.i = save
                    break
                node = ("rep", node, lo, hi)
            else:
                break
        return node

    def _atom(self):
        ch = self._next()
        if ch == "(":
            if self._peek() == "?":
                self._next()
                nxt = self._peek()
                if nxt == ":":
                    self._next()
                else:
                    raise ValueError(f"unsupported group (?{nxt}")
            node = self._alt()
            if self._peek() != ")":
                raise ValueError("unbalanced (")
            self._next()
            return node
        if ch == "[":
            return ("lit", self._char_class())
        if ch == ".":
            return ("lit", CharClass(_DOT))
        if ch == "\\":
            return ("lit", self._escape())
        if ch in "*+?{":
            raise ValueError(f"dangling quantifier {ch!r}")
        if ch in ("^", "$"):
            # Whole-string anchoring is implicit; treat as empty.
            return ("empty",)
        return ("lit", CharClass(((ord(ch), ord(ch)),)))

    def _escape(self) -> CharClass:
        ch = self._next()
        if ch in _ESCAPES:
            return CharClass(_ESCAPES[ch])
        if ch in _CTRL:
            c = _CTRL[ch]
            return CharClass(((ord(c), ord(c)),))
        if ch in ("x", "u"):
            n = 2 if ch == "x" else 4
            hexs = self.p[self.i:self.i + n]
            self.i += n
            cp = int(hexs, 16)
            return CharClass(((cp, cp),))
        return CharClass(((ord(ch), ord(ch)),))

    def _char_class(self) -> CharClass:
        negated = False
        if self._peek() == "^":
            self._next()
            negated = True
        ranges = []
        first = True
        while True:
            ch = self._peek()
            if ch is None:
                raise ValueError("unbalanced [")
            if ch == "]" and not first:
                self._next()
                break
            first = False
            ch = self._next()
            if ch == "\\":
                nxt = self._peek()
                if nxt in _ESCAPES:
                    self._next()
                    ranges.extend(_ESCAPES[nxt])
                    continue
                cc = self._escape()
                lo = cc.ranges[0][0]
            else:
                lo = ord(ch)
            if self._peek() == "-" and self.i + 1 < len(self.p) \
                    and self.p[self.i + 1] != "]":
                self._next()
                hi_ch = self._next()
                if hi_ch == "\\":
                    hi = self._escape().ranges[0][0]
                else:
                    hi = ord(hi_ch)
                ranges.append((lo, hi))
            else:
                ranges.append((lo, lo))
        merged = _merge(ranges)
        return CharClass(_negate(merged) if negated else merged)


def _build_nfa(pattern: str) -> _NFA:
    ast = _Parser(pattern).parse()
    nfa = _NFA()

    def emit(node, entry: int) -> int:
        """Wire `node` from state `entry`; return its exit state."""
        kind = node[0]
        if kind == "empty":
            return entry
        if kind == "lit":
            nxt = nfa._new()
            nfa.trans[entry].append((node[1], nxt))
            return nxt
        if kind == "cat":
            cur = entry
            for child in node[1]:
                cur = emit(child, cur)
            return cur
        if kind == "alt":
            exit_ = nfa._new()
            for child in node[1]:
                b_in = nfa._new()
                nfa.eps[entry].append(b_in)
                b_out = emit(child, b_in)
                nfa.eps[b_out].append(exit_)
            return exit_
        if kind == "rep":
            _, child, lo, hi = node
            cur = entry
            for _ in range(lo):
                cur = emit(child, cur)
            if hi is None:  # unbounded tail: loop
                loop_in = nfa._new()
                nfa.eps[cur].append(loop_in)
                body_out = emit(child, loop_in)
                nfa.eps[body_out].append(loop_in)
                exit_ = nfa._new()
                nfa.eps[loop_in].append(exit_)
                return exit_
            for _ in range(hi - lo):  # optional copies
                nxt = emit(child, cur)
                nfa.eps[cur].append(nxt)
                cur = nxt
            return cur
        raise AssertionError(kind)

    nfa.accept = emit(ast, nfa.start)
    return nfa


# --------------------------------------------------------------------------
# Lazy DFA over the NFA, with liveness (can-reach-accept) precomputed.
# --------------------------------------------------------------------------


class RegexFSM:
    def __init__(self, pattern: str):
        self.pattern = pattern
        nfa = _build_nfa(pattern)
        self._nfa = nfa
        # Backward eps+char reachability to the accept state → live set.
        n = len(nfa.eps)
        preds: list[list[int]] = [[] for _ in range(n)]
        for s in range(n):
            for t in nfa.eps[s]:
                preds[t].append(s)
            for _, t in nfa.trans[s]:
                preds[t].append(s)
        live = set()
        stack = [nfa.accept]
        while stack:
            s = stack.pop()
            if s in live:
                continue
            live.add(s)
            stack.extend(preds[s])
        self._live = live

        self._dfa_sets: list[frozenset] = []
        self._dfa_ids: dict[frozenset, int] = {}
        self._accepting: list[bool] = []
        self._step_cache: dict[tuple[int, int], Optional[int]] = {}
        self.start = self._dfa_state(self._closure({nfa.start}))

    def _closure(self, states: set) -> frozenset:
        stack = list(states)
        out = set(states)
        while stack:
            s = stack.pop()
            for t in self._nfa.eps[s]:
                if t not in out:
                    out.add(t)
                    stack.append(t)
        return frozenset(s for s in out if s in self._live)

    def _dfa_state(self, closed: frozenset) -> Optional[int]:
        if not closed:
            return None
        sid = self._dfa_ids.get(closed)
        if sid is None:
            sid = len(self._dfa_sets)
            self._dfa_ids[closed] = sid
            self._dfa_sets.append(closed)
            self._accepting.append(self._nfa.accept in closed)
        return sid

    def step(self, sid: int, cp: int) -> Optional[int]:
        """DFA transition on codepoint; None = dead."""
        key = (sid, cp)
        hit = self._step_cache.get(key, _MISS)
        if hit is not _MISS:
            return hit
        nxt = set()
        for s in self._dfa_sets[sid]:
            for cc, t in self._nfa.trans[s]:
                if cc.matches(cp):
                    nxt.add(t)
        res = self._dfa_state(self._closure(nxt)) if nxt else None
        self._step_cache[key] = res
        return res

    def step_str(self, sid: Optional[int], text: str) -> Optional[int]:
        for ch in text:
            if sid is None:
                return None
            sid = self.step(sid, ord(ch))
        return sid

    def is_accepting(self, sid: int) -> bool:
        return self._accepting[sid]

    def fullmatch(self, text: str) -> bool:
        sid = self.step_str(self.start, text)
        return sid is not None and self._accepting[sid]


_MISS = object()


# --------------------------------------------------------------------------
# Vocab trie (shared per tokenizer) and the token-level grammar.
# --------------------------------------------------------------------------


class _TrieNode:
    __slots__ = ("children", "token_ids")

    def __init__(self):
        self.children: dict[str, "_TrieNode"] = {}
        self.token_ids: list[int] = []


def build_vocab_trie(tokenizer) -> _TrieNode:
    """Character trie over the decoded string of every vocab token.
    Tokens that don't decode to clean text (byte-fallback fragments,
    specials) are excluded — they can never appear in a constrained
    output."""
    root = _TrieNode()
    vocab_size = getattr(tokenizer, "vocab_size", None)
    inner = getattr(tokenizer, "tokenizer", tokenizer)
    if vocab_size is None:
        vocab_size = getattr(inner, "vocab_size", 0)
    special = set()
    for attr in ("bos_token_id", "eos_token_id", "pad_token_id",
                 "unk_token_id"):
        tid = getattr(inner, attr, None)
        if tid is not None:
            special.add(tid)
    all_special = getattr(inner, "all_special_ids", None)
    if all_special:
        special.update(all_special)

    decode = inner.decode
    for tid in range(vocab_size):
        if tid in special:
            continue
        try:
            text = decode([tid])
        except Exception:  # noqa: BLE001
            continue
        if not text or "�" in text:
            continue
        node = root
        for ch in text:
            node = node.children.setdefault(ch, _TrieNode())
        node.token_ids.append(tid)
    return root


_TRIE_CACHE: dict[int, _TrieNode] = {}


def vocab_trie_for(tokenizer) -> _TrieNode:
    key = id(getattr(tokenizer, "tokenizer", tokenizer))
    trie = _TRIE_CACHE.get(key)
    if trie is None:
        trie = build_vocab_trie(tokenizer)
        _TRIE_CACHE[key] = trie
    return trie


class RegexGrammar:
    """Token-level grammar over a compiled regex FSM. Same surface as
    structured_output.CompiledGrammar (initial_state / allowed_tokens /
    advance / is_exhausted); states are DFA ids (int)."""

    def __init__(self, pattern: str, tokenizer,
                 eos_token_id: Optional[int]):
        self.fsm = RegexFSM(pattern)
        self.trie = vocab_trie_for(tokenizer)
        self.eos_token_id = eos_token_id
        self._mask_cache: dict[int, set[int]] = {}
        # (dfa_state, token) -> next dfa_state, filled by the mask walk.
        self._adv: dict[tuple[int, int], int] = {}
        # token id -> decoded text (for advance() on cache miss)
        self._tok_text: dict[int, str] = {}
        self._index_tok_text(self.trie, "")

    def _index_tok_text(self, node: _TrieNode, prefix: str) -> None:
        for tid in node.token_ids:
            self._tok_text[tid] = prefix
        for ch, child in node.children.items():
            self._index_tok_text(child, prefix + ch)

    def initial_state(self) -> int:
        return self.fsm.start

    def allowed_tokens(self, state: int) -> set[int]:
        cached = self._mask_cache.get(state)
        if cached is not None:
            return cached
        allowed: set[int] = set()
        stack = [(self.trie, state)]
        adv = self._adv
        while stack:
            node, sid = stack.pop()
            for ch, child in node.children.items():
                nxt = self.fsm.step(sid, ord(ch))
                if nxt is None:
                    continue
                for tid in child.token_ids:
                    allowed.add(tid)
                    adv[(state, tid)] = nxt
                if child.children:
                    stack.append((child, nxt))
        if self.fsm.is_accepting(state) and self.eos_token_id is not None:
            allowed.add(self.eos_token_id)
        self._mask_cache[state] = allowed
        return allowed

    def advance(self, state: int, token: int) -> Optional[int]:
        nxt = self._adv.get((state, token))
        if nxt is not None:
            return nxt
        text = self._tok_text.get(token)
        if text is None:  # EOS / special / unknown → leaves the grammar
            return None
        return self.fsm.step_str(state, text)

    def is_exhausted(self, state: int) -> bool:
        allowed = self.allowed_tokens(state)
        return not (allowed - {self.eos_token_id})
