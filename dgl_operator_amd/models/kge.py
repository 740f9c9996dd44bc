"""Knowledge-graph embedding model (DGL-KE equivalent).

Single-process variant of the reference's KE training stack
(/root/reference/examples/DGL-KE/hotfix/kvclient.py + train_pytorch): entity
and relation embeddings trained with chunked negative sampling,
self-adversarial logsigmoid loss, and row-sparse Adagrad. In the distributed
setting, entity embeddings live in the sharded KVStore
(dgl_operator_amd.distributed.kvstore) instead of this module.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import get_score_func, kge_loss
from ..ops.adagrad import sparse_adagrad_update


class KGEModel(nn.Module):
    def __init__(
        self,
        num_entities: int,
        num_relations: int,
        hidden_dim: int = 400,
        score_func: str = "ComplEx",
        gamma: float = 143.0,
        device: str | torch.device = "cpu",
    ):
        super().__init__()
        self.num_entities = num_entities
        self.num_relations = num_relations
        self.hidden_dim = hidden_dim
        emb_init = (gamma + 2.0) / hidden_dim
        self.emb_init = emb_init
        self.score = get_score_func(score_func, gamma=gamma, emb_init=emb_init)
        rel_dim = hidden_dim
        if score_func == "RotatE":
            rel_dim = hidden_dim // 2  # one phase per complex dimension
        elif score_func == "RESCAL":
            rel_dim = hidden_dim * hidden_dim
        elif score_func == "TransR":
            rel_dim = hidden_dim * (1 + hidden_dim)
        dev = torch.device(device)
        self.entity_emb = torch.empty(num_entities, hidden_dim, device=dev)
        self.relation_emb = torch.empty(num_relations, rel_dim, device=dev)
        self.entity_state = torch.zeros(num_entities, device=dev)
        self.relation_state = torch.zeros(num_relations, device=dev)
        nn.init.uniform_(self.entity_emb, -emb_init, emb_init)
        nn.init.uniform_(self.relation_emb, -emb_init, emb_init)

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
    ) -> float:
        """One KE step with local embeddings (single GPU / CPU path)."""
        B = heads.numel()
        num_chunk = neg_entities.shape[0]
        assert B == num_chunk * chunk_size
        h = self.entity_emb[heads].requires_grad_(True)
        r = self.relation_emb[rels].requires_grad_(True)
        t = self.entity_emb[tails].requires_grad_(True)
        n = self.entity_emb[neg_entities.reshape(-1)].requires_grad_(True)

        pos = self.score.edge(h, r, t)
        hc = (t if neg_head else h).view(num_chunk, chunk_size, -1)
        rc = r.view(num_chunk, chunk_size, -1)
        nc = n.view(num_chunk, neg_entities.shape[1], -1)
        neg = self.score.neg(hc, rc, nc, neg_head=neg_head)
        loss = kge_loss(pos, neg, adversarial_temperature)
        loss.backward()

        with torch.no_grad():
            ent_ids = torch.cat([heads, tails, neg_entities.reshape(-1)])
            ent_grad = torch.cat([h.grad, t.grad, n.grad.view(-1, self.hidden_dim)])
            sparse_adagrad_update(
                self.entity_emb, self.entity_state, ent_ids, ent_grad, lr
            )
            sparse_adagrad_update(
                self.relation_emb, self.relation_state, rels, r.grad, lr
            )
        return float(loss.detach())

    def predict(self, heads, rels, tails) -> torch.Tensor:
        with torch.no_grad():
            return self.score.edge(
                self.entity_emb[heads], self.relation_emb[rels], self.entity_emb[tails]
            )

    def save(self, path: str):
        torch.save(
            {
                "entity_emb": self.entity_emb.cpu(),
                "relation_emb": self.relation_emb.cpu(),
                "entity_state": self.entity_state.cpu(),
                "relation_state": self.relation_state.cpu(),
                "hidden_dim": self.hidden_dim,
                "score_func": self.score.name,
            },
            path,
        )
