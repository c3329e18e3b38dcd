"""Expert-parallel token exchange (MoE all-to-all over xGMI;
BASELINE.json config 5: Mixtral-8x7B TP+EP across 8 MI355X).

Mixtral's routed FFN shards experts across the EP group (8 experts over up
to 8 GPUs: each rank holds n_experts/ep_size experts' full weights —
288 GB HBM easily fits them).  Tokens travel to their experts' owner ranks
and back with RCCL all_to_all_single over xGMI — the bandwidth-optimal
pattern for point-to-point links: every rank sends only the rows each peer
actually needs, no broadcast amplification.

On gloo (CPU tests) all_to_all is unsupported; the same exchange runs via
all_gather_object — semantically identical, so world_size=2 CPU tests
cover the routing/combine logic the GPU runs over RCCL.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def all_to_all_rows(chunks: list[torch.Tensor], group=None) -> list[torch.Tensor]:
    """chunks[r] = rows to send to rank r ([n_r, H]); returns the rows
    received from each rank, same order.  Row counts may be ragged."""
    world = dist.get_world_size(group)
    backend = dist.get_backend(group)
    if backend == "nccl":
        H = chunks[0].shape[1]
        dev = chunks[0].device
        send_counts = torch.tensor([c.shape[0] for c in chunks],
                                   dtype=torch.int64, device=dev)
        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts, group=group)
        send = torch.cat(chunks, dim=0)
        recv = torch.empty(int(recv_counts.sum()), H, dtype=send.dtype,
                           device=dev)
        dist.all_to_all_single(
            recv, send,
            output_split_sizes=recv_counts.tolist(),
            input_split_sizes=send_counts.tolist(), group=group)
        out, off = [], 0
        for n in recv_counts.tolist():
            out.append(recv[off:off + n])
            off += n
        return out
    # gloo emulation (CPU tests): object gather of the per-peer chunks
    rank = dist.get_rank(group)
    gathered: list = [None] * world
    dist.all_gather_object(gathered, [c.cpu() for c in chunks], group=group)
    return [gathered[src][rank].to(chunks[0].device) for src in range(world)]


def all_gather_rows(local: torch.Tensor, group=None) -> torch.Tensor:
    """Concatenate ragged [n_r, H] row blocks from every rank, rank order."""
    world = dist.get_world_size(group)
    backend = dist.get_backend(group)
    if backend == "nccl":
        dev = local.device
        n = torch.tensor([local.shape[0]], dtype=torch.int64, device=dev)
        counts = [torch.empty_like(n) for _ in range(world)]
        dist.all_gather(counts, n, group=group)
        counts = [int(c.item()) for c in counts]
        mx = max(counts)
        pad = torch.zeros(mx, local.shape[1], dtype=local.dtype, device=dev)
        pad[:local.shape[0]] = local
        outs = [torch.empty_like(pad) for _ in range(world)]
        dist.all_gather(outs, pad, group=group)
        return torch.cat([o[:c] for o, c in zip(outs, counts)], dim=0)
    gathered: list = [None] * world
    dist.all_gather_object(gathered, local.cpu(), group=group)
    return torch.cat([g.to(local.device) for g in gathered], dim=0)


class ExpertDispatch:
    """One MoE layer's token routing across the EP group.

    dispatch(): group each token-choice by owning rank, exchange.
    combine(): reverse exchange, weighted scatter-add into [T, H] output.
    """

    def __init__(self, n_experts: int, ep_size: int, ep_rank: int,
                 group=None):
        assert n_experts % ep_size == 0
        self.n_experts = n_experts
        self.ep_size = ep_size
        self.ep_rank = ep_rank
        self.group = group
        self.per_rank = n_experts // ep_size

    def owner(self, expert: int) -> int:
        return expert // self.per_rank

    def run(self, h: torch.Tensor, top_idx: torch.Tensor,
            top_w: torch.Tensor, expert_fn) -> torch.Tensor:
        """h [T, H]; top_idx/top_w [T, K] routing choices.
        expert_fn(local_expert_id, rows) -> rows' FFN output.
        Returns [T, H] combined output."""
        T, H = h.shape
        K = top_idx.shape[1]
        flat_tok = torch.arange(T, device=h.device).repeat_interleave(K)
        flat_exp = top_idx.reshape(-1)
        flat_w = top_w.reshape(-1)
        if self.ep_size == 1:
            out = torch.zeros(T, H, dtype=torch.float32, device=h.device)
            for e in range(self.n_experts):
                sel = (flat_exp == e).nonzero(as_tuple=True)[0]
                if sel.numel() == 0:
                    continue
                toks = flat_tok[sel]
                y = expert_fn(e, h[toks])
                out.index_add_(0, toks,
                               y.float() * flat_w[sel].float().unsqueeze(1))
            return out.to(h.dtype)

        owner = flat_exp // self.per_rank
        send_chunks, send_meta = [], []
        for r in range(self.ep_size):
            sel = (owner == r).nonzero(as_tuple=True)[0]
            # sort by expert so the remote compute is grouped per expert
            sel = sel[flat_exp[sel].argsort(stable=True)]
            send_meta.append(sel)
            # row = [hidden | expert_id] so one exchange carries both
            rows = torch.cat([h[flat_tok[sel]].float(),
                              flat_exp[sel].float().unsqueeze(1)], dim=1)
            send_chunks.append(rows)
        recv = all_to_all_rows(send_chunks, self.group)

        # compute local experts on every received chunk
        results = []
        base = self.ep_rank * self.per_rank
        for rows in recv:
            hrows, erows = rows[:, :H], rows[:, H].long()
            y = torch.zeros(rows.shape[0], H, dtype=torch.float32,
                            device=h.device)
            for le in range(self.per_rank):
                sel = (erows == base + le).nonzero(as_tuple=True)[0]
                if sel.numel():
                    y[sel] = expert_fn(base + le,
                                       hrows[sel].to(h.dtype)).float()
            results.append(y)
        back = all_to_all_rows(results, self.group)

        out = torch.zeros(T, H, dtype=torch.float32, device=h.device)
        for sel, y in zip(send_meta, back):
            if sel.numel():
                out.index_add_(0, flat_tok[sel],
                               y * flat_w[sel].float().unsqueeze(1))
        return out.to(h.dtype)
