"""Fused Mixture-of-Experts layer (role of the reference's
vllm/model_executor/layers/fused_moe/layer.py FusedMoE).

Two sharding modes over the same forward:

- TP (default): every rank holds all experts with the intermediate dim
  sharded 1/tp (gate/up halves sharded separately inside w13).
- EP (enable_expert_parallel): experts sharded across ranks at full
  intermediate width; tokens are replicated in SPMD TP serving, so each
  rank computes its local experts for every token and the combine is
  the same RCCL all-reduce (the naive AgRs manager of the reference's
  all2all.py — DeepEP-style dispatch/combine lands later).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd import ops
from vllm_amd.layers.linear import ReplicatedLinear
from vllm_amd.parallel.state import (
    get_tp_rank,
    get_tp_world_size,
    tensor_model_parallel_all_reduce,
)


class FusedMoE(nn.Module):

    def __init__(
        self,
        num_experts: int,
        top_k: int,
        hidden_size: int,
        intermediate_size: int,
        renormalize: bool = True,
        activation: str = "silu",
        dtype: torch.dtype = None,
        enable_expert_parallel: bool = False,
        eplb_window: int = 0,
    ):
        super().__init__()
        from vllm_amd.parallel.state import get_ep_group

        tp = get_tp_world_size()
        self.ep_size = 1
        self.ep_rank = 0
        if enable_expert_parallel and tp > 1:
            ep = get_ep_group()
            self.ep_size = ep.world_size
            self.ep_rank = ep.rank_in_group
        if self.ep_size > 1:
            assert num_experts % self.ep_size == 0, (
                f"{num_experts} experts not divisible by ep {self.ep_size}")
            self.num_local_experts = num_experts // self.ep_size
            self.expert_lo = self.ep_rank * self.num_local_experts
            self.i_shard = intermediate_size  # full width per local expert
        else:
            assert intermediate_size % tp == 0, (
                f"moe intermediate {intermediate_size} not divisible by "
                f"tp {tp}")
            self.num_local_experts = num_experts
            self.expert_lo = 0
            self.i_shard = intermediate_size // tp
        self.num_experts = num_experts
        self.top_k = top_k
        self.renormalize = renormalize
        self.activation = activation
        self.gate = ReplicatedLinear(hidden_size, num_experts, bias=False,
                                     dtype=dtype)
        # TP: [E, 2*I/tp, H]; EP: [E/ep, 2*I, H].
        self.w13 = nn.Parameter(
            torch.empty(self.num_local_experts, 2 * self.i_shard,
                        hidden_size, dtype=dtype),
            requires_grad=False,
        )
        self.w2 = nn.Parameter(
            torch.empty(self.num_local_experts, hidden_size, self.i_shard,
                        dtype=dtype),
            requires_grad=False,
        )
        # EPLB (role of the reference's vllm/distributed/eplb,
        # eplb_state.py:220): dynamic expert->rank placement. owner_of /
        # slot_of are device lookup tensors consulted by the EP routing
        # mask; rebalance() re-packs experts by EWMA token load and
        # moves weights IN PLACE (hipGraphs stay valid). All ranks see
        # identical router outputs (replicated activations), so every
        # rank computes the same plan with no control-plane sync — only
        # the weight broadcasts communicate.
        self.eplb_window = eplb_window if self.ep_size > 1 else 0
        if self.ep_size > 1:
            n = self.num_local_experts
            self.assignment = [e // n for e in range(num_experts)]
            self.local_slots = list(range(self.expert_lo,
                                          self.expert_lo + n))
            self.register_buffer(
                "owner_of",
                torch.tensor(self.assignment, dtype=torch.int64),
                persistent=False)
            self.register_buffer(
                "slot_of",
                torch.tensor([e % n for e in range(num_experts)],
                             dtype=torch.int64),
                persistent=False)
            # Device-side token-load accumulator: the += in forward is
            # hipGraph-capturable and keeps accumulating across graph
            # REPLAYS; the runner triggers rebalance() between steps
            # (collectives and the D2H read cannot run inside a capture).
            self.register_buffer("eplb_load",
                                 torch.zeros(num_experts,
                                             dtype=torch.int64),
                                 persistent=False)
            self._eplb_ewma = None
        # Fragment-major weight copies for the HIP grouped GEMM (built
        # lazily on the first GPU forward; freed by invalidate_shuffled
        # when weights change). Costs a second copy of the expert
        # weights — the MI355X trade: 288 GB HBM buys coalesced,
        # barrier-free weight streams. Opt-in via VLLM_AMD_MOE_SHUF=1
        # (the staged-LDS kernel measured faster at decode batch>=256).
        self._w13_shuf = None
        self._w2_shuf = None

    # -- EPLB ----------------------------------------------------------
    def _eplb_observe(self, topk_ids) -> None:
        self.eplb_load += torch.bincount(topk_ids.reshape(-1),
                                         minlength=self.num_experts)

    def _plan_assignment(self, load) -> list[int]:
        """Greedy balanced packing: experts by load (desc), each to the
        least-loaded rank with free slots. Deterministic (ties broken by
        expert id / rank id) so every rank computes the same plan."""
        w = self.ep_size
        cap = self.num_local_experts
        order = sorted(range(self.num_experts),
                       key=lambda e: (-load[e], e))
        rank_load = [0.0] * w
        rank_cnt = [0] * w
        plan = [0] * self.num_experts
        for e in order:
            r = min((r for r in range(w) if rank_cnt[r] < cap),
                    key=lambda r: (rank_load[r], r))
            plan[e] = r
            rank_load[r] += load[e]
            rank_cnt[r] += 1
        return plan

    @torch.no_grad()
    def rebalance(self) -> None:
        """Re-pack experts onto ranks by EWMA load and move weights.
        Collective: every rank must call this at the same step (SPMD
        lockstep guarantees it when triggered by the shared counter)."""
        import torch.distributed as dist

        from vllm_amd.parallel.state import get_ep_group

        ep = get_ep_group()
        cur = self.eplb_load.detach().cpu().double().numpy()
        if self._eplb_ewma is None:
            self._eplb_ewma = cur.copy()
        else:
            self._eplb_ewma = 0.5 * self._eplb_ewma + 0.5 * cur
        self.eplb_load.zero_()
        plan = self._plan_assignment(self._eplb_ewma.tolist())
        if plan == self.assignment:
            return
        # Per-rank slot order: owned experts ascending by global id.
        new_slots = [e for e in range(self.num_experts)
                     if plan[e] == self.ep_rank]
        slot_index = {}
        for r in range(self.ep_size):
            owned = [e for e in range(self.num_experts) if plan[e] == r]
            for s, e in enumerate(owned):
                slot_index[e] = s
        # Stash current local weights (sources for broadcasts AND for
        # intra-rank slot moves).
        stash13 = self.w13.data.clone()
        stash2 = self.w2.data.clone()
        old_slot = {e: s for s, e in enumerate(self.local_slots)}
        buf13 = torch.empty_like(self.w13.data[0])
        buf2 = torch.empty_like(self.w2.data[0])
        for e in range(self.num_experts):
            src = self.assignment[e]
            dst = plan[e]
            if src == dst and e in old_slot and slot_index[e] == old_slot[e]:
                continue  # unchanged owner and slot
            if src == dst:
                # intra-rank slot move only
                if dst == self.ep_rank:
                    self.w13.data[slot_index[e]].copy_(stash13[old_slot[e]])
                    self.w2.data[slot_index[e]].copy_(stash2[old_slot[e]])
                continue
            if self.ep_rank == src:
                buf13.copy_(stash13[old_slot[e]])
                buf2.copy_(stash2[old_slot[e]])
            src_global = (ep.group_ranks[src]
                          if hasattr(ep, "group_ranks") else src)
            dist.broadcast(buf13, src=src_global, group=ep.device_group)
            dist.broadcast(buf2, src=src_global, group=ep.device_group)
            if self.ep_rank == dst:
                self.w13.data[slot_index[e]].copy_(buf13)
                self.w2.data[slot_index[e]].copy_(buf2)
        self.assignment = plan
        self.local_slots = new_slots
        self.owner_of.copy_(torch.tensor(plan, dtype=torch.int64))
        slot_list = [slot_index[e] for e in range(self.num_experts)]
        self.slot_of.copy_(torch.tensor(slot_list, dtype=torch.int64))
        self.refresh_shuffled()

    def refresh_shuffled(self) -> None:
        """Call after in-place weight updates (RL update_weights):
        re-shuffles INTO THE EXISTING storage so captured hipGraphs that
        reference the fragment-major tensors stay valid."""
        if self._w13_shuf is None:
            return
        from vllm_amd.ops import get_backend
        backend = get_backend(self.w13.device)
        self._w13_shuf.copy_(backend.moe_shuffle_weights(self.w13.data))
        self._w2_shuf.copy_(backend.moe_shuffle_weights(self.w2.data))

    def _maybe_shuffled(self, hidden):
        import os

        # Fragment-major streaming is opt-in (VLLM_AMD_MOE_SHUF=1): on
        # MI355X the staged-LDS grouped GEMM measured faster at decode
        # batch >= 256 (the barrier-free stream wins only at tiny
        # batches; see profiles/r02_summary.md).
        if (not hidden.is_cuda
                or os.environ.get("VLLM_AMD_MOE_SHUF", "0") != "1"):
            return None, None
        if self._w13_shuf is None:
            from vllm_amd.ops import get_backend
            backend = get_backend(hidden.device)
            shuffle = getattr(backend, "moe_shuffle_weights", None)
            if shuffle is None or not backend._moe_hip_ok(
                    hidden, self.w13, self.w2, self.activation):
                return None, None
            self._w13_shuf = shuffle(self.w13.data)
            self._w2_shuf = shuffle(self.w2.data)
        return self._w13_shuf, self._w2_shuf

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        router_logits = self.gate(hidden)
        topk_weights, topk_ids = ops.topk_softmax(
            router_logits, self.top_k, renormalize=self.renormalize
        )
        out = self.run_experts(hidden, topk_weights, topk_ids)
        if get_tp_world_size() > 1:
            out = tensor_model_parallel_all_reduce(out)
        return out

    def run_experts(self, hidden, topk_weights, topk_ids) -> torch.Tensor:
        """Local expert computation. EP mode: activations are replicated
        across ranks (SPMD TP serving), so each rank COMPACTS to the
        (token, slot) pairs routed to its local experts — the per-rank
        expert GEMM token count is already T*k/ep, identical to what an
        all-to-all dispatch would compute — and the caller's all-reduce
        completes the combine. An a2a dispatch/combine only REDUCES
        traffic once the residual stream itself is sequence-sharded
        through the MoE block (SP covers llama-family dense blocks;
        SP-MoE is the tracked follow-on), so the replicated regime keeps
        the simpler, deterministic collective."""
        if self.ep_size > 1:
            if self.eplb_window > 0:
                self._eplb_observe(topk_ids)
            owner = self.owner_of.to(topk_ids.device)
            slot_map = self.slot_of.to(topk_ids.device)
            mask = owner[topk_ids] == self.ep_rank
            slot_tok, slot_k = mask.nonzero(as_tuple=True)
            if slot_tok.numel() == 0:
                return torch.zeros_like(hidden)
            sel_ids = slot_map[topk_ids[slot_tok, slot_k]].unsqueeze(1)
            sel_w = topk_weights[slot_tok, slot_k].unsqueeze(1)
            w13s, w2s = self._maybe_shuffled(hidden)
            y = ops.fused_moe(
                hidden[slot_tok], self.w13, self.w2, sel_w, sel_ids,
                activation=self.activation, w13_shuf=w13s, w2_shuf=w2s,
            )
            out = torch.zeros(hidden.shape, dtype=torch.float32,
                              device=hidden.device)
            out.index_add_(0, slot_tok, y.float())
            return out.to(hidden.dtype)
        w13s, w2s = self._maybe_shuffled(hidden)
        return ops.fused_moe(
            hidden, self.w13, self.w2, topk_weights, topk_ids,
            activation=self.activation, w13_shuf=w13s, w2_shuf=w2s,
        )

    def load_full_weights(self, w1_full, w3_full, w2_full) -> None:
        """Shard full expert weights onto this rank: w1/w3 [E, I, H]
        (gate / up), w2 [E, H, I]."""
        i = self.i_shard
        if self.ep_size > 1:
            lo, hi = self.expert_lo, self.expert_lo + self.num_local_experts
            self.w13.data[:, :i] = w1_full[lo:hi]
            self.w13.data[:, i:] = w3_full[lo:hi]
            self.w2.data.copy_(w2_full[lo:hi])
            return
        tp, r = get_tp_world_size(), get_tp_rank()
        self.w13.data[:, :i] = w1_full[:, r * i:(r + 1) * i]
        self.w13.data[:, i:] = w3_full[:, r * i:(r + 1) * i]
        self.w2.data.copy_(w2_full[:, :, r * i:(r + 1) * i])
