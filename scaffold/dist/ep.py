"""Expert parallelism (SURVEY §2.4 additive layer): a top-1-routed MoE
feed-forward with the EXPERTS sharded across the EP group.

Data flow (dense-dispatch formulation, exact and fully differentiable):
  1. all_gather the group's token shards (differentiable: backward
     reduce-scatters dx to the owner ranks);
  2. every rank evaluates the replicated gate over ALL tokens and runs
     only ITS experts on the tokens routed to them, scattering the
     gate-weighted outputs into a zero buffer;
  3. a differentiable all_reduce sums the per-rank partial outputs
     (each token was computed on exactly one rank), and each rank keeps
     its own token slice.
Expert weight gradients stay rank-local (each rank owns its experts);
gate gradients are averaged by the caller's DDP like any replicated
parameter.  The production dispatch would exchange only the routed
tokens with an all_to_all — same math, sparser transport (NOTES.md).
"""

from __future__ import annotations


import torch
import torch.nn as nn


class ExpertParallelMoE(nn.Module):
    """Top-1 MoE FFN; `n_experts` must be divisible by `world`."""

    def __init__(self, d_model: int, d_ff: int, n_experts: int, world: int,
                 rank: int, group=None, dispatch: str = "alltoall",
                 init_seed: int = 0, top_k: int = 1, act: str = "gelu"):
        super().__init__()
        if n_experts % world:
            raise ValueError(f"n_experts {n_experts} not divisible by world {world}")
        if dispatch not in ("alltoall", "dense"):
            raise ValueError(f"unknown dispatch {dispatch!r}")
        if not 1 <= top_k <= n_experts:
            raise ValueError(f"top_k {top_k} out of range 1..{n_experts}")
        if act not in ("gelu", "swiglu"):
            raise ValueError(f"unknown expert act {act!r}")
        self.top_k = top_k
        self.act = act
        self.world = world
        self.rank = rank
        self.group = group
        self.dispatch = dispatch
        self.n_experts = n_experts
        self.e_local = n_experts // world
        self.e0 = rank * self.e_local
        self.gate = nn.Linear(d_model, n_experts, bias=False)  # replicated
        # expert weights draw from PER-GLOBAL-EXPERT generators, not the
        # ambient (rank-seeded) stream: expert g is bit-identical no
        # matter which rank owns it or what the world size is, so an
        # EP-sharded model == the single-process model weight-for-weight
        # (and the ambient stream position stays world-invariant too)
        def _init(shape, gid):
            g = torch.Generator().manual_seed(init_seed * 1000003 + gid)
            return nn.Parameter(torch.randn(*shape, generator=g) * 0.02)

        # swiglu experts pack gate|up rows in one matrix (the same
        # layout as the dense Llama block's w_gateup)
        up_rows = 2 * d_ff if act == "swiglu" else d_ff
        self.w_up: nn.ParameterList = nn.ParameterList(
            _init((up_rows, d_model), self.e0 + i)
            for i in range(self.e_local))
        self.w_down: nn.ParameterList = nn.ParameterList(
            _init((d_model, d_ff), 10**6 + self.e0 + i)
            for i in range(self.e_local))

    def _zero_tie(self, i_local: int) -> torch.Tensor:
        # an expert that received no tokens this step must still appear
        # in the graph: a bucketed-DDP reducer waits on EVERY param's
        # grad hook, so an absent expert would stall the all-reduce
        # (grad = exact zero either way)
        return 0.0 * (self.w_up[i_local].flatten()[0]
                      + self.w_down[i_local].flatten()[0])

    def _expert(self, i_local: int, x: torch.Tensor) -> torch.Tensor:
        if self.act == "swiglu":
            gu = x @ self.w_up[i_local].t()
            F_ = gu.shape[-1] // 2
            h = torch.nn.functional.silu(gu[..., :F_]) * gu[..., F_:]
        else:
            h = torch.nn.functional.gelu(x @ self.w_up[i_local].t(),
                                         approximate="tanh")
        return h @ self.w_down[i_local].t()

    def forward(self, x_local: torch.Tensor) -> torch.Tensor:
        import torch.distributed as dist

        if (self.dispatch == "alltoall" and self.world > 1
                and dist.is_initialized()):
            return self._forward_alltoall(x_local)
        return self._forward_dense(x_local)

    def _forward_alltoall(self, x_local: torch.Tensor) -> torch.Tensor:
        """Sparse dispatch: exchange only the ROUTED tokens (SURVEY
        §2.4 / NOTES r1 planned form).  Transport per rank is its own T
        tokens out + back, vs the dense form's (world-1)*T all_gather +
        world*T all_reduce — and it uses all_to_all, the collective
        xGMI serves with world-1 concurrent p2p links.

        Gating runs on LOCAL tokens; the gate weight travels nowhere
        (the prob multiplies the RETURNED expert output, keeping the
        gate gradient rank-local).  Expert ids ride a separate non-grad
        all_to_all with the same splits."""
        import torch.distributed as dist
        from torch.distributed import nn as dist_nn

        probs = torch.softmax(self.gate(x_local), dim=-1)
        # top-k: each token travels to its k chosen experts (k-fold
        # transport — the standard cost of Mixtral-style routing); the
        # returned outputs are mixed with renormalized gate weights
        top_p, top_e = probs.topk(self.top_k, dim=-1)        # (T, k)
        top_p = top_p / top_p.sum(dim=-1, keepdim=True)
        T = x_local.shape[0]
        k = self.top_k
        flat_e = top_e.reshape(-1)                           # (T*k,)
        flat_p = top_p.reshape(-1)
        src_tok = (torch.arange(T * k, device=x_local.device) // k)
        dest = flat_e // self.e_local                        # owner rank

        order = torch.argsort(dest, stable=True)
        inv = torch.argsort(order)
        x_s = x_local[src_tok[order]]
        e_s = flat_e[order]

        send_counts = torch.bincount(dest, minlength=self.world)
        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts, group=self.group)
        send_l = send_counts.tolist()
        recv_l = recv_counts.tolist()

        n_recv = int(sum(recv_l))
        x_recv = torch.empty(n_recv, x_local.shape[1], dtype=x_local.dtype,
                             device=x_local.device)
        x_recv = dist_nn.functional.all_to_all_single(
            x_recv, x_s.contiguous(), output_split_sizes=recv_l,
            input_split_sizes=send_l, group=self.group)
        e_recv = torch.empty(n_recv, dtype=e_s.dtype, device=e_s.device)
        dist.all_to_all_single(e_recv, e_s.contiguous(),
                               output_split_sizes=recv_l,
                               input_split_sizes=send_l, group=self.group)

        # x_recv * 0, NOT zeros_like: when ZERO tokens land on this
        # rank's experts there is no index_add, and a graph-free zero
        # would let autograd PRUNE the feature-a2a backward on THIS
        # rank while the peers (whose experts did get tokens) still run
        # it — divergent collective sequences that corrupt/abort the
        # group (found via per-rank collective tracing when a collapsed
        # gate routed everything to one rank)
        y_recv = x_recv * 0.0
        for i in range(self.e_local):
            sel = (e_recv == self.e0 + i).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                y_recv = y_recv + self._zero_tie(i)
                continue
            y_recv = y_recv.index_add(0, sel, self._expert(i, x_recv[sel]))

        y_back = torch.empty_like(x_s)
        y_back = dist_nn.functional.all_to_all_single(
            y_back, y_recv.contiguous(), output_split_sizes=send_l,
            input_split_sizes=recv_l, group=self.group)
        mixed = flat_p.unsqueeze(-1) * y_back[inv]           # (T*k, d)
        return mixed.view(T, k, -1).sum(dim=1)

    def _forward_dense(self, x_local: torch.Tensor) -> torch.Tensor:
        import torch.distributed as dist

        T = x_local.shape[0]
        if self.world > 1 and dist.is_initialized():
            from torch.distributed import nn as dist_nn

            x_all = torch.cat(
                list(dist_nn.functional.all_gather(x_local, group=self.group)),
                dim=0)
        else:
            x_all = x_local

        probs = torch.softmax(self.gate(x_all), dim=-1)      # (T_all, E)
        # top-k routing: each token mixes its k best experts' outputs,
        # weighted by their (renormalized) gate probabilities
        top_p, top_e = probs.topk(self.top_k, dim=-1)        # (T_all, k)
        top_p = top_p / top_p.sum(dim=-1, keepdim=True)

        y_partial = x_all * 0.0   # keep the graph tie (see alltoall path)
        for i in range(self.e_local):
            sel_t, sel_k = (top_e == self.e0 + i).nonzero(as_tuple=True)
            if sel_t.numel() == 0:
                y_partial = y_partial + self._zero_tie(i)
                continue
            out = self._expert(i, x_all[sel_t]) \
                * top_p[sel_t, sel_k].unsqueeze(-1)
            y_partial = y_partial.index_add(0, sel_t, out)

        if self.world > 1 and dist.is_initialized():
            from torch.distributed import nn as dist_nn

            y_all = dist_nn.functional.all_reduce(y_partial, group=self.group)
            return y_all[self.rank * T:(self.rank + 1) * T]
        return y_partial


def mark_ep_checkpoint_identities(model, world: int, rank: int) -> None:
    """Tag sharded expert params with global checkpoint identities and
    build the model-level manifest map the checkpoint layer consumes
    (see scaffold/ckpt/sharded.py): LOCAL slot i on this rank holds
    GLOBAL expert e0+i, so shards key by global id and rank 0 can write
    a complete manifest for experts it does not hold."""
    model._pinned_manifest = {}
    for li, blk in enumerate(model.blocks):
        moe = getattr(blk, "moe", None)
        if moe is None:
            continue
        for kind, plist in (("w_up", moe.w_up), ("w_down", moe.w_down)):
            for i, p_ in enumerate(plist):
                gid = moe.e0 + i
                p_._ckpt_name = f"blocks.{li}.moe.{kind}.g{gid}"
                p_._ckpt_owner = rank
            for gid in range(moe.n_experts):
                model._pinned_manifest[
                    f"blocks.{li}.moe.{kind}.g{gid}"] = (
                    gid // moe.e_local, list(plist[0].shape))
