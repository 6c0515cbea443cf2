"""Draft-model speculative decoding (role of the reference's
vllm/v1/spec_decode/draft_model.py): an independent small causal LM
proposes k tokens per step; the target verifies them with the exact
rejection rule in model_runner.

Runner-side state: the draft model runs through the SAME forward-context
machinery as the target (paged KV + attention_unified), against its own
dedicated KV pool with trivial per-request allocation (draft KV is tiny
— capacity is preallocated per seat, no eviction). The observe/propose
protocol mirrors EAGLE's but is token-conditioned, not
feature-conditioned:

  observe(rid, start, tokens)  commit final tokens at [start, start+T)
  propose(rid, last_token)     feed the newest token speculatively, then
                               k greedy self-feeding forwards; the
                               speculative cache slots roll back (paged
                               writes are overwritten on the next
                               commit, so rollback = reset the length).
"""

from __future__ import annotations

import torch

from vllm_amd.worker.forward_context import (
    AttentionMetadata,
    ForwardContext,
    set_forward_context,
)


class _DraftReq:
    __slots__ = ("block_ids", "len")

    def __init__(self, block_ids):
        self.block_ids = block_ids
        self.len = 0


class DraftModelRunnerSide:

    BLOCK = 16

    def __init__(self, model, spec, k: int, max_len: int, max_seqs: int,
                 dtype, device):
        assert not spec.is_mla, "MLA draft models not supported"
        self.model = model
        self.spec = spec
        self.k = k
        self.device = device
        self.dtype = dtype
        bs = self.BLOCK
        self.blocks_per_seq = -(-(max_len + k + 1) // bs)
        num_blocks = self.blocks_per_seq * max_seqs + 1
        from vllm_amd.parallel.state import get_tp_world_size
        kv_heads = spec.num_kv_heads_per_rank(get_tp_world_size())
        self.kv_caches = [
            torch.zeros(2, num_blocks, kv_heads, bs, spec.head_dim,
                        dtype=dtype, device=device)
            for _ in range(spec.num_layers)
        ]
        self._free = [list(range(1 + i * self.blocks_per_seq,
                                 1 + (i + 1) * self.blocks_per_seq))
                      for i in range(max_seqs)]
        self._req: dict[str, _DraftReq] = {}

    def free(self, rid: str) -> None:
        st = self._req.pop(rid, None)
        if st is not None:
            self._free.append(st.block_ids)

    def _state(self, rid: str) -> _DraftReq:
        st = self._req.get(rid)
        if st is None:
            if not self._free:
                raise RuntimeError("draft KV pool exhausted")
            st = _DraftReq(self._free.pop())
            self._req[rid] = st
        return st

    def _forward(self, st: _DraftReq, tokens: list[int], start: int,
                 decode: bool) -> torch.Tensor:
        """Run the draft over `tokens` at positions [start, start+T);
        returns last-position hidden. KV rows land in st's blocks."""
        T = len(tokens)
        dev = self.device
        bs = self.BLOCK
        ids = torch.tensor(tokens, dtype=torch.int64, device=dev)
        pos = torch.arange(start, start + T, dtype=torch.int64, device=dev)
        bt = torch.tensor([st.block_ids], dtype=torch.int32, device=dev)
        blk = pos // bs
        slot = (bt[0, blk].to(torch.int64) * bs + pos % bs)
        meta = AttentionMetadata(
            query_start_loc=torch.tensor([0, T], dtype=torch.int32,
                                         device=dev),
            seq_lens=torch.tensor([start + T], dtype=torch.int32,
                                  device=dev),
            block_table=bt,
            slot_mapping=slot,
            num_reqs=1,
            num_actual_tokens=T,
            max_query_len=1 if decode else T,
            max_seq_len=start + T,
            num_decodes=1 if (decode and T == 1) else 0,
        )
        ctx = ForwardContext(attn_metadata=meta, kv_caches=self.kv_caches)
        with set_forward_context(ctx):
            hidden = self.model(ids, pos)
        return hidden[-1]

    @torch.inference_mode()
    def observe(self, rid: str, start: int, tokens: list[int]) -> None:
        if not tokens:
            return
        st = self._state(rid)
        if start > st.len:
            # Gap (should not happen: every final position is observed);
            # resync from scratch next time.
            st.len = 0
            return
        self._forward(st, tokens, start, decode=False)
        st.len = start + len(tokens)

    @torch.inference_mode()
    def propose(self, rid: str, last_token: int) -> list[int]:
        st = self._req.get(rid)
        if st is None or st.len == 0:
            return []
        if st.len + self.k + 1 > self.blocks_per_seq * self.BLOCK:
            return []
        base = st.len
        drafts: list[int] = []
        tok = int(last_token)
        for _ in range(self.k):
            h = self._forward(st, [tok], st.len, decode=True)
            st.len += 1
            tok = int(self.model.compute_logits(h.unsqueeze(0)).argmax())
            drafts.append(tok)
        st.len = base  # roll back speculative slots
        return drafts


class DraftModelProposer:
    """Scheduler-side marker: drafts come from the runner
    (ModelRunnerOutput.draft_token_ids), like EAGLE/Medusa."""

    model_based = True
