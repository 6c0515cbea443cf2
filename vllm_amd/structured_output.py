"""Structured (grammar-constrained) output.

Role of the reference's vllm/v1/structured_output/: a per-request
compiled grammar produces a per-step allowed-token mask that the sampler
applies before sampling. Round-1 backend: `guided_choice` — the output
must be exactly one of N strings (compiled to a token trie). Grammar'd
requests run with synchronous scheduling (the mask for step N+1 depends
on step N's token).
"""

from __future__ import annotations

from typing import Optional


class TrieNode:
    __slots__ = ("children", "terminal")

    def __init__(self):
        self.children: dict[int, "TrieNode"] = {}
        self.terminal = False


class CompiledGrammar:
    """Token-trie grammar for a fixed choice set."""

    def __init__(self, token_sequences: list[list[int]],
                 eos_token_id: Optional[int]):
        self.root = TrieNode()
        self.eos_token_id = eos_token_id
        for seq in token_sequences:
            node = self.root
            for tok in seq:
                node = node.children.setdefault(tok, TrieNode())
            node.terminal = True

    def initial_state(self) -> TrieNode:
        return self.root

    def allowed_tokens(self, state: TrieNode) -> set[int]:
        allowed = set(state.children)
        if state.terminal and self.eos_token_id is not None:
            allowed.add(self.eos_token_id)
        return allowed

    def advance(self, state: TrieNode, token: int) -> Optional[TrieNode]:
        """Next state, or None when the token ends/leaves the grammar."""
        return state.children.get(token)

    def is_exhausted(self, state: TrieNode) -> bool:
        return not state.children


def compile_choice_grammar(choices: list[str], tokenizer,
                           eos_token_id: Optional[int]) -> CompiledGrammar:
    seqs = []
    for choice in choices:
        ids = tokenizer.encode(choice)
        # Strip a leading BOS if the tokenizer added one.
        bos = getattr(tokenizer, "bos_token_id", None)
        if bos is None and hasattr(tokenizer, "tokenizer"):
            bos = getattr(tokenizer.tokenizer, "bos_token_id", None)
        if bos is not None and ids and ids[0] == bos:
            ids = ids[1:]
        if not ids:
            raise ValueError(f"choice {choice!r} tokenizes to nothing")
        seqs.append(ids)
    return CompiledGrammar(seqs, eos_token_id)
