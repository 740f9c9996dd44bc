"""Distributed KGE training — the dglke_dist_train equivalent.

Entity AND relation embeddings live in ShardedEmbeddings (one shard per GPU
over xGMI; the reference partitions relations too — hotfix/kvclient.py:56-72).
Each step: pull rows, score + self-adversarial loss with autograd over the
pulled rows, push grads to their owning shards where fused sparse Adagrad
applies them.

Reference call stack being replaced: SURVEY.md §3.4 (kvclient ->
dist_train_test -> KVClient.pull/push -> KGEServer sparse Adagrad).
The head/tail corruption alternates per step like the reference's
NewBidirectionalOneShotIterator (hotfix/sampler.py:823-875).
"""
from __future__ import annotations

from typing import Tuple

import torch

from ..ops import get_score_func, kge_loss
from .kvstore import ShardedEmbedding


class DistKGEModel:
    def __init__(
        self,
        num_entities: int,
        num_relations: int,
        hidden_dim: int,
        score_func: str = "TransE_l2",
        gamma: float = 12.0,
        rank: int = 0,
        world_size: int = 1,
        device="cpu",
        seed: int = 0,
        entity_boundaries=None,
        relation_boundaries=None,
    ):
        self.num_entities = num_entities
        self.hidden_dim = hidden_dim
        emb_init = (gamma + 2.0) / hidden_dim
        self.score = get_score_func(score_func, gamma=gamma, emb_init=emb_init)
        self.entities = ShardedEmbedding(
            num_entities, hidden_dim, world_size, rank, device=device,
            init_range=emb_init, seed=seed, boundaries=entity_boundaries,
        )
        rel_dim = hidden_dim
        if score_func == "RotatE":
            rel_dim = hidden_dim // 2
        elif score_func == "RESCAL":
            rel_dim = hidden_dim * hidden_dim
        elif score_func == "TransR":
            rel_dim = hidden_dim * (1 + hidden_dim)
        # relations sharded too, like the reference's relation partition
        # (hotfix/kvclient.py:56-72 round-robin); ranges work equally here
        self.relations = ShardedEmbedding(
            num_relations, rel_dim, world_size, rank, device=device,
            init_range=emb_init, seed=seed + 7,
            boundaries=relation_boundaries,
        )

    def train_step(
        self,
        heads: torch.Tensor,
        rels: torch.Tensor,
        tails: torch.Tensor,
        neg_entities: torch.Tensor,  # [num_chunk, neg]
        chunk_size: int,
        lr: float,
        neg_head: bool = False,
        adversarial_temperature: float = 1.0,
        regularization_coef: float = 0.0,
        regularization_norm: int = 3,
    ) -> float:
        B = heads.numel()
        num_chunk, n_neg = neg_entities.shape
        assert B == num_chunk * chunk_size
        dim = self.hidden_dim
        # ONE entity pull/push round per step: h, t and the negatives share a
        # single alltoallv (the reference pulls them separately through the
        # kvstore — dis_kvstore.py:818-902)
        ent_ids = torch.cat([heads, tails, neg_entities.reshape(-1)])
        rows = self.entities.pull(ent_ids).requires_grad_(True)
        h = rows[:B]
        t = rows[B : 2 * B]
        n = rows[2 * B :]
        r = self.relations.pull(rels).requires_grad_(True)

        pos = self.score.edge(h, r, t)
        hc = (t if neg_head else h).reshape(num_chunk, chunk_size, dim)
        rc = r.view(num_chunk, chunk_size, -1)
        nc = n.reshape(num_chunk, n_neg, dim)
        neg = self.score.neg(hc, rc, nc, neg_head=neg_head)
        loss = kge_loss(pos, neg, adversarial_temperature)
        if regularization_coef:
            # DGL-KE's Lp regularization of the batch's entity embeddings
            # (the reference runs dglke_dist_train with
            # --regularization_coef 1e-9, dglkerun:301; norm p=3 is the
            # dglke default). Gradients flow into the same push.
            p = regularization_norm
            loss = loss + regularization_coef * (
                rows.abs().pow(p).sum() / max(rows.shape[0], 1))
        loss.backward()

        with torch.no_grad():
            self.entities.push_grad(ent_ids, rows.grad, lr)
            self.relations.push_grad(rels, r.grad, lr)
        return float(loss.detach())


class KGEdgeSampler:
    """Chunked negative edge sampler (K8). Yields
    (heads, rels, tails, neg_entities, neg_head) with head/tail corruption
    alternating per step. Negatives are uniform entity draws shared per chunk
    (DGL-KE's chunked corruption shape: batch/chunk_size chunks x neg)."""

    def __init__(
        self,
        triples: Tuple[torch.Tensor, torch.Tensor, torch.Tensor],
        num_entities: int,
        batch_size: int = 1024,
        neg_sample_size: int = 256,
        chunk_size: int = 64,
        seed: int = 0,
        device="cpu",
    ):
        self.h, self.r, self.t = (x.to(device) for x in triples)
        self.num_entities = num_entities
        self.batch_size = batch_size
        self.neg = neg_sample_size
        self.chunk = chunk_size
        self.gen = torch.Generator(device=device)
        self.gen.manual_seed(seed)
        self.step = 0
        self.device = device

    def next_batch(self):
        E = self.h.numel()
        idx = torch.randint(0, E, (self.batch_size,), generator=self.gen,
                            device=self.device)
        num_chunk = self.batch_size // self.chunk
        negs = torch.randint(
            0, self.num_entities, (num_chunk, self.neg), generator=self.gen,
            device=self.device,
        )
        neg_head = self.step % 2 == 1
        self.step += 1
        return self.h[idx], self.r[idx], self.t[idx], negs, neg_head


@torch.no_grad()
def evaluate_kge(
    model: DistKGEModel,
    heads: torch.Tensor,
    rels: torch.Tensor,
    tails: torch.Tensor,
    batch_size: int = 128,
    hits: Tuple[int, ...] = (1, 3, 10),
    corrupt: str = "tail",
    filter_triples: "Tuple[torch.Tensor, torch.Tensor, torch.Tensor] | None" = None,
):
    """Link-prediction metrics (MRR, MR, Hits@K) — the dglke eval protocol
    (reference EvalSampler/EvalDataset, hotfix/sampler.py:514-821): rank the
    true entity among ALL entities. With sharded entity tables each rank
    scores the candidates it OWNS and the global rank is the all-reduced
    count of higher-scoring candidates + 1.

    ``filter_triples`` switches to the FILTERED setting: known triples
    (h, r, t') with t' != t are excluded from the ranking by subtracting the
    known competitors that outscore the true entity (every rank is assumed
    to hold the full triple list, as in this repo's synthetic KGs).

    Collective contract: every ``pull`` below is an alltoallv, so ALL RANKS
    MUST PASS IDENTICAL (heads, rels, tails) AND filter sets — asserted up
    front. The filtered rescoring issues exactly ONE batched pull per eval
    batch (never a data-dependent number of collectives per triple).

    Tie handling is optimistic ('>'): the true entity itself sits in the
    candidate shard, and its score there comes from the chunked ``neg``
    kernel while ``true_score`` comes from ``edge`` — bitwise equality
    between the two paths is not guaranteed, so a '>=' (pessimistic) count
    could penalize the truth against itself.
    """
    import torch.distributed as dist

    device = heads.device
    if dist.is_available() and dist.is_initialized() and \
            dist.get_world_size() > 1:
        dev = model.entities.local.device \
            if dist.get_backend() == "nccl" else torch.device("cpu")
        sig = torch.tensor(
            [float(heads.numel()), float(heads.sum()), float(rels.sum()),
             float(tails.sum())], dtype=torch.float64, device=dev)
        lo, hi = sig.clone(), sig.clone()
        dist.all_reduce(lo, op=dist.ReduceOp.MIN)
        dist.all_reduce(hi, op=dist.ReduceOp.MAX)
        assert torch.equal(lo, hi), (
            "evaluate_kge: all ranks must evaluate the SAME triples "
            "(every pull is a collective; divergent loops deadlock)")
    rank_sum = 0.0
    rr_sum = 0.0
    hit_counts = {k: 0.0 for k in hits}
    n_total = heads.numel()
    shard = model.entities.local  # [M, D] owned candidate entities
    M = shard.shape[0]
    known = None
    if filter_triples is not None:
        from collections import defaultdict

        kh, kr, kt = filter_triples
        known = defaultdict(list)
        if corrupt == "tail":
            for a, b, c in zip(kh.tolist(), kr.tolist(), kt.tolist()):
                known[(a, b)].append(c)
        else:
            for a, b, c in zip(kh.tolist(), kr.tolist(), kt.tolist()):
                known[(c, b)].append(a)
    for s in range(0, n_total, batch_size):
        hh = heads[s : s + batch_size]
        rr = rels[s : s + batch_size]
        tt = tails[s : s + batch_size]
        B = hh.numel()
        h = model.entities.pull(hh)
        r = model.relations.pull(rr)
        t = model.entities.pull(tt)
        true_score = model.score.edge(h, r, t)  # [B]
        if corrupt == "tail":
            cand = model.score.neg(
                h.view(1, B, -1), r.view(1, B, -1), shard.view(1, M, -1),
                neg_head=False,
            )[0]  # [B, M]
        else:
            cand = model.score.neg(
                t.view(1, B, -1), r.view(1, B, -1), shard.view(1, M, -1),
                neg_head=True,
            )[0]
        higher = (cand > true_score.unsqueeze(1)).sum(1).double()
        if dist.is_available() and dist.is_initialized():
            dist.all_reduce(higher)
        if known is not None:
            # filtered: remove known competitors that outscored the truth.
            # ONE batched pull for the whole eval batch (identical on all
            # ranks by the contract asserted above), then local rescoring.
            adj = torch.zeros_like(higher)
            cand_lists = []
            for i in range(B):
                key = (int(hh[i]), int(rr[i]))
                excl = int(tt[i] if corrupt == "tail" else hh[i])
                cand_lists.append(
                    [e for e in known.get(key, ()) if e != excl])
            flat = [e for lst in cand_lists for e in lst]
            if flat:
                ke_all = model.entities.pull(
                    torch.as_tensor(flat, device=device))
                off = 0
                for i, lst in enumerate(cand_lists):
                    if not lst:
                        continue
                    ke = ke_all[off : off + len(lst)]
                    off += len(lst)
                    if corrupt == "tail":
                        sc = model.score.edge(
                            h[i].expand(len(lst), -1),
                            r[i].expand(len(lst), -1), ke)
                    else:
                        sc = model.score.edge(
                            ke, r[i].expand(len(lst), -1),
                            t[i].expand(len(lst), -1))
                    adj[i] = (sc > true_score[i]).sum()
            higher = (higher - adj).clamp(min=0)
        ranks = higher + 1.0
        rank_sum += float(ranks.sum())
        rr_sum += float((1.0 / ranks).sum())
        for k in hits:
            hit_counts[k] += float((ranks <= k).sum())
    return {
        "MR": rank_sum / n_total,
        "MRR": rr_sum / n_total,
        **{f"Hits@{k}": hit_counts[k] / n_total for k in hits},
    }


def relation_partition_order(
    rels_of_triples: torch.Tensor, num_relations: int, num_parts: int,
    mode: str = "soft",
):
    """Relation->part assignment balancing per-part TRIPLE counts — parity
    with DGL-KE's relation partitioners (reference hotfix/sampler.py:
    SoftRelationPartition :32-148, BalancedRelationPartition :150,
    RandomPartition :256-290). Returns (new_of_old, boundaries): relations are
    RELABELED so each part owns one contiguous id range (what
    ShardedEmbedding shards by); heavy relations go first so the greedy
    least-loaded packing balances like the reference's.
    """
    counts = torch.bincount(rels_of_triples, minlength=num_relations)
    if mode == "random":
        order = torch.randperm(
            num_relations, generator=torch.Generator().manual_seed(0)
        )
        assign = torch.arange(num_relations)[order] % num_parts
        assign = assign[torch.argsort(order)]  # assignment per original id
    else:  # soft / balanced: greedy least-loaded over descending frequency
        order = torch.argsort(counts, descending=True)
        loads = [0] * num_parts
        assign = torch.empty(num_relations, dtype=torch.int64)
        for rid in order.tolist():
            p = min(range(num_parts), key=lambda q: loads[q])
            assign[rid] = p
            loads[p] += int(counts[rid])
    perm = torch.argsort(assign, stable=True)  # new id -> old id
    new_of_old = torch.empty_like(perm)
    new_of_old[perm] = torch.arange(num_relations)
    sizes = torch.bincount(assign, minlength=num_parts)
    boundaries = [0]
    for s in sizes.tolist():
        boundaries.append(boundaries[-1] + s)
    return new_of_old, boundaries
