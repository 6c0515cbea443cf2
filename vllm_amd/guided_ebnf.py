"""Recursive GBNF grammars as a character-level pushdown automaton.

Completes guided_grammar.py: non-recursive grammars still compile to a
regex for the lazy-DFA engine (cheaper masks), while grammars with rule
cycles — balanced parentheses, nested JSON, expression languages — run
on this PDA (role of the reference's xgrammar backend; the matching
strategy is the llama.cpp grammar-sampler one: a SET of symbol stacks
advanced character by character, rule references expanded on demand).

Symbol encoding: a rule reference is its non-negative rule id; a
character class is -(class_id+1). A "stack" is a tuple of symbols still
to match (leftmost first); a PDA state is an interned frozenset of
stacks whose top symbol is a character class (or the empty stack =
accepting). Left-recursive rules make that expansion diverge and are
rejected at compile time with a clear error (same limitation as
llama.cpp; the reference's xgrammar also rewrites rather than supports
them directly).

Exposes the same surface as guided_regex.RegexFSM (start / step /
step_str / is_accepting), so EbnfGrammar reuses RegexGrammar's
vocab-trie mask walk unchanged.
"""

from __future__ import annotations

from typing import Optional

from vllm_amd.guided_grammar import (
    GrammarError,
    _ESCAPE,
    _NAME_RE,
    _parse_rules,
)
from vllm_amd.guided_regex import RegexGrammar, vocab_trie_for

# Bounds: pathological grammars/paths terminate instead of exploding.
_MAX_EXPAND = 4096    # distinct stacks touched while expanding one state
_MAX_STACK_DEPTH = 2048  # symbols per stack (nesting depth bound)


class _Compiler:
    """GBNF text -> numbered rules of alternates of symbols."""

    def __init__(self, grammar: str):
        named = _parse_rules(grammar)
        self.rule_ids: dict[str, int] = {}
        self.rules: list[list[list[int]]] = []
        self.classes: list[tuple[bool, tuple[tuple[int, int], ...]]] = []
        self._class_ids: dict = {}
        for name in named:  # two-pass: ids first, then bodies
            self.rule_ids[name] = len(self.rules)
            self.rules.append([])
        for name, body in named.items():
            alts = self._parse_alternates(body)
            self.rules[self.rule_ids[name]] = alts
        self.root = self.rule_ids["root"]
        self._check_left_recursion()

    def _check_left_recursion(self) -> None:
        """Reject rules that can derive themselves without consuming a
        character (the stack-set expansion would diverge on them)."""
        nullable: set[int] = set()
        changed = True
        while changed:
            changed = False
            for rid, alts in enumerate(self.rules):
                if rid in nullable:
                    continue
                for alt in alts:
                    if all(s >= 0 and s in nullable for s in alt):
                        nullable.add(rid)
                        changed = True
                        break
        # Edge rid -> s when s can be the leftmost unconsumed rule.
        edges: list[set[int]] = [set() for _ in self.rules]
        for rid, alts in enumerate(self.rules):
            for alt in alts:
                for s in alt:
                    if s < 0:
                        break  # a character class consumes input
                    edges[rid].add(s)
                    if s not in nullable:
                        break
        color = [0] * len(self.rules)  # 0 new, 1 on stack, 2 done

        def visit(rid: int) -> None:
            color[rid] = 1
            for s in edges[rid]:
                if color[s] == 1:
                    names = {v: k for k, v in self.rule_ids.items()}
                    raise GrammarError(
                        f"left-recursive rule "
                        f"{names.get(s, f'#{s}')!r}: rewrite with "
                        "right recursion or repetition")
                if color[s] == 0:
                    visit(s)
            color[rid] = 2

        for rid in range(len(self.rules)):
            if color[rid] == 0:
                visit(rid)

    # -- symbol constructors -------------------------------------------
    def _class_sym(self, negated: bool,
                   ranges: tuple[tuple[int, int], ...]) -> int:
        key = (negated, ranges)
        cid = self._class_ids.get(key)
        if cid is None:
            cid = len(self.classes)
            self.classes.append(key)
            self._class_ids[key] = cid
        return -(cid + 1)

    def _char_sym(self, ch: str) -> int:
        return self._class_sym(False, ((ord(ch), ord(ch)),))

    def _new_rule(self, alts: list[list[int]]) -> int:
        rid = len(self.rules)
        self.rules.append(alts)
        return rid

    # -- repetition desugaring -----------------------------------------
    def _repeat(self, syms: list[int], op: str) -> list[int]:
        """Wrap the symbols of one atom per * + ? (fresh helper rules)."""
        if op == "?":
            return [self._new_rule([syms, []])]
        star = self._new_rule([])  # S ::= syms S | ""
        self.rules[star] = [syms + [star], []]
        if op == "*":
            return [star]
        return syms + [star]  # +: one mandatory copy then the star

    # -- body parsing --------------------------------------------------
    def _parse_alternates(self, body: str) -> list[list[int]]:
        atoms: list[list[list[int]]] = [[]]  # per-alt atom list
        i, n = 0, len(body)

        def push_atom(syms: list[int]) -> None:
            atoms[-1].append(syms)

        while i < n:
            ch = body[i]
            if ch.isspace():
                i += 1
            elif ch == "|":
                atoms.append([])
                i += 1
            elif ch == '"':
                j = i + 1
                lit = []
                while j < n and body[j] != '"':
                    if body[j] == "\\" and j + 1 < n:
                        lit.append(_ESCAPE.get(body[j + 1], body[j + 1]))
                        j += 2
                    else:
                        lit.append(body[j])
                        j += 1
                if j >= n:
                    raise GrammarError(
                        f"unterminated string in: {body!r}")
                # One atom per literal so a postfix op repeats the WHOLE
                # literal (llama.cpp semantics).
                push_atom([self._char_sym(c) for c in lit])
                i = j + 1
            elif ch == "[":
                j, sym = self._parse_class(body, i)
                push_atom([sym])
                i = j
            elif ch == "(":
                depth, j = 1, i + 1
                while j < n and depth:
                    if body[j] == "(":
                        depth += 1
                    elif body[j] == ")":
                        depth -= 1
                    elif body[j] == '"':
                        j += 1
                        while j < n and body[j] != '"':
                            j += 2 if body[j] == "\\" else 1
                    j += 1
                if depth:
                    raise GrammarError(f"unbalanced parens in: {body!r}")
                inner = self._parse_alternates(body[i + 1:j - 1])
                push_atom([self._new_rule(inner)])
                i = j
            elif ch in "*+?":
                if not atoms[-1]:
                    raise GrammarError(f"dangling {ch!r} in: {body!r}")
                atoms[-1][-1] = self._repeat(atoms[-1][-1], ch)
                i += 1
            else:
                m = _NAME_RE.match(body, i)
                if not m:
                    raise GrammarError(f"unexpected {ch!r} in: {body!r}")
                name = m.group(0)
                if name not in self.rule_ids:
                    raise GrammarError(f"undefined rule {name!r}")
                push_atom([self.rule_ids[name]])
                i = m.end()
        return [[s for atom in alist for s in atom] for alist in atoms]

    def _parse_class(self, body: str, i: int) -> tuple[int, int]:
        n = len(body)
        j = i + 1
        negated = body[j:j + 1] == "^"
        if negated:
            j += 1
        ranges: list[tuple[int, int]] = []
        while j < n and body[j] != "]":
            if body[j] == "\\" and j + 1 < n:
                lo = ord(_ESCAPE.get(body[j + 1], body[j + 1]))
                j += 2
            else:
                lo = ord(body[j])
                j += 1
            hi = lo
            if body[j:j + 1] == "-" and body[j + 1:j + 2] not in ("]", ""):
                j += 1
                if body[j] == "\\" and j + 1 < n:
                    hi = ord(_ESCAPE.get(body[j + 1], body[j + 1]))
                    j += 2
                else:
                    hi = ord(body[j])
                    j += 1
            ranges.append((lo, hi))
        if j >= n:
            raise GrammarError(f"unterminated class in: {body!r}")
        return j + 1, self._class_sym(negated, tuple(ranges))


class EbnfFSM:
    """PDA with the RegexFSM stepping surface; states are interned ids."""

    def __init__(self, grammar: str):
        c = _Compiler(grammar)
        self.rules = c.rules
        self.classes = c.classes
        self._ids: dict[frozenset, int] = {}
        self._configs: list[frozenset] = []
        cfg = self._expand({(c.root,)})
        if not cfg:
            raise GrammarError("grammar matches nothing")
        self.start = self._intern(cfg)
        if not self._configs[self.start]:
            raise GrammarError("grammar matches nothing")

    def _intern(self, cfg: frozenset) -> int:
        sid = self._ids.get(cfg)
        if sid is None:
            sid = len(self._configs)
            self._configs.append(cfg)
            self._ids[cfg] = sid
        return sid

    def _expand(self, stacks) -> frozenset:
        """Expand rule-reference tops until every stack starts with a
        character class (or is empty). Diverging expansion = left
        recursion -> compile-time GrammarError; over-deep stacks are
        dropped (bounded memory) rather than grown without limit."""
        out = set()
        seen = set()
        work = list(stacks)
        while work:
            st = work.pop()
            if st in seen or len(st) > _MAX_STACK_DEPTH:
                continue
            seen.add(st)
            if len(seen) > _MAX_EXPAND:
                raise GrammarError(
                    "grammar expansion diverged (left-recursive rule?)")
            if not st or st[0] < 0:
                out.add(st)
                continue
            for alt in self.rules[st[0]]:
                work.append(tuple(alt) + st[1:])
        return frozenset(out)

    def _match(self, sym: int, cp: int) -> bool:
        negated, ranges = self.classes[-sym - 1]
        hit = any(lo <= cp <= hi for lo, hi in ranges)
        return hit != negated

    def step(self, sid: int, cp: int) -> Optional[int]:
        nxt = {st[1:] for st in self._configs[sid]
               if st and self._match(st[0], cp)}
        if not nxt:
            return None
        cfg = self._expand(nxt)
        if not cfg:
            return None
        return self._intern(cfg)

    def step_str(self, sid: Optional[int], text: str) -> Optional[int]:
        for ch in text:
            if sid is None:
                return None
            sid = self.step(sid, ord(ch))
        return sid

    def is_accepting(self, sid: int) -> bool:
        return any(not st for st in self._configs[sid])


class EbnfGrammar(RegexGrammar):
    """Recursive-grammar token masks: RegexGrammar's vocab-trie walk
    over the PDA instead of the regex DFA."""

    def __init__(self, grammar: str, tokenizer,
                 eos_token_id: Optional[int]):
        self.fsm = EbnfFSM(grammar)
        self.trie = vocab_trie_for(tokenizer)
        self.eos_token_id = eos_token_id
        self._mask_cache = {}
        self._adv = {}
        self._tok_text = {}
        self._index_tok_text(self.trie, "")
