"""guided_grammar: GBNF-style EBNF compiled to a regex for the existing
lazy-DFA token-mask engine (guided_regex.py).

Role of the reference's grammar backend (xgrammar EBNF). Scope: the
NON-RECURSIVE subset — rules are inlined bottom-up, so any rule cycle is
rejected with a clear error (recursive grammars need a pushdown matcher;
tracked for a later round). Supported syntax (llama.cpp GBNF):

    root  ::= "yes" | "no" | digits
    digits ::= [0-9]+
    ws    ::= [ \\t]*

terminals "..." (with \\ escapes), char classes [...], grouping (...),
postfix * + ?, alternation |, concatenation by juxtaposition, and
`#` comments.
"""

from __future__ import annotations

import re

_RULE_RE = re.compile(r"^\s*([A-Za-z_][A-Za-z0-9_-]*)\s*::=\s*(.*)$")
_NAME_RE = re.compile(r"[A-Za-z_][A-Za-z0-9_-]*")
_ESCAPE = {"n": "\n", "t": "\t", "r": "\r", '"': '"', "\\": "\\"}


class GrammarError(ValueError):
    pass


def _strip_comments(line: str) -> str:
    out = []
    in_str = in_cls = False
    i = 0
    while i < len(line):
        ch = line[i]
        if ch == "\\" and (in_str or in_cls):
            out.append(line[i:i + 2])
            i += 2
            continue
        if ch == '"' and not in_cls:
            in_str = not in_str
        elif ch == "[" and not in_str:
            in_cls = True
        elif ch == "]" and not in_str:
            in_cls = False
        elif ch == "#" and not in_str and not in_cls:
            break
        out.append(ch)
        i += 1
    return "".join(out)


def _parse_rules(grammar: str) -> dict[str, str]:
    """Split the grammar into {rule_name: body} (continuation lines
    attach to the previous rule)."""
    rules: dict[str, str] = {}
    current = None
    for raw in grammar.splitlines():
        line = _strip_comments(raw).rstrip()
        if not line.strip():
            continue
        m = _RULE_RE.match(line)
        if m:
            current = m.group(1)
            rules[current] = m.group(2).strip()
        elif current is not None:
            rules[current] += " " + line.strip()
        else:
            raise GrammarError(f"grammar line outside a rule: {raw!r}")
    if "root" not in rules:
        raise GrammarError("grammar must define a 'root' rule")
    return rules


def _body_to_regex(body: str, rules: dict[str, str],
                   stack: tuple[str, ...]) -> str:
    """Translate one rule body, inlining referenced rules (cycle check
    via `stack`)."""
    out = []
    i = 0
    n = len(body)
    while i < n:
        ch = body[i]
        if ch.isspace():
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
                raise GrammarError(f"unterminated string in: {body!r}")
            out.append("(?:" + re.escape("".join(lit)) + ")")
            i = j + 1
        elif ch == "[":
            j = i + 1
            while j < n and body[j] != "]":
                if body[j] == "\\":
                    j += 1
                j += 1
            if j >= n:
                raise GrammarError(f"unterminated class in: {body!r}")
            out.append(body[i:j + 1])
            i = j + 1
        elif ch == "(":
            depth = 1
            j = i + 1
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
            inner = _body_to_regex(body[i + 1:j - 1], rules, stack)
            out.append("(?:" + inner + ")")
            i = j
        elif ch in "*+?":
            if not out:
                raise GrammarError(f"dangling {ch!r} in: {body!r}")
            out[-1] = out[-1] + ch
            i += 1
        elif ch == "|":
            out.append("|")
            i += 1
        else:
            m = _NAME_RE.match(body, i)
            if not m:
                raise GrammarError(f"unexpected {ch!r} in: {body!r}")
            name = m.group(0)
            if name not in rules:
                raise GrammarError(f"undefined rule {name!r}")
            if name in stack:
                raise GrammarError(
                    f"recursive rule {name!r} (cycle {' -> '.join(stack)}"
                    f" -> {name}); only non-recursive grammars compile "
                    "to a regex")
            inner = _body_to_regex(rules[name], rules, stack + (name,))
            out.append("(?:" + inner + ")")
            i = m.end()
    return "".join(out)


def grammar_to_regex(grammar: str) -> str:
    """Compile a GBNF-style grammar string to a regex accepted by
    guided_regex.RegexGrammar."""
    rules = _parse_rules(grammar)
    return _body_to_regex(rules["root"], rules, ("root",))
