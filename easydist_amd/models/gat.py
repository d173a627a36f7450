"""Graph attention network (dense adjacency) — reference bench model.

reference: benchmark/torch/model/gat.py and bench_case.py:21-24
(GAT num_node 4096, in_feature 12288). Dense-adjacency formulation: the
attention score matrix is [N, N], which on MI355X is one MFMA GEMM +
masked softmax — the right formulation for a 4096-node benchmark graph.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class GATLayer(nn.Module):
    def __init__(self, in_dim, out_dim):
        super().__init__()
        self.fc = nn.Linear(in_dim, out_dim, bias=False)
        self.attn_l = nn.Parameter(torch.empty(out_dim))
        self.attn_r = nn.Parameter(torch.empty(out_dim))
        nn.init.normal_(self.attn_l, std=0.1)
        nn.init.normal_(self.attn_r, std=0.1)

    def forward(self, x, adj):
        h = self.fc(x)                              # [N, D]
        el = (h * self.attn_l).sum(-1)              # [N]
        er = (h * self.attn_r).sum(-1)
        scores = F.leaky_relu(el.unsqueeze(1) + er.unsqueeze(0), 0.2)
        scores = scores.masked_fill(adj == 0, float("-inf"))
        alpha = torch.softmax(scores, dim=-1)
        alpha = torch.nan_to_num(alpha, nan=0.0)    # isolated nodes
        return torch.matmul(alpha, h)


class GAT(nn.Module):
    def __init__(self, in_dim=12288, hidden=512, n_classes=64):
        super().__init__()
        self.l1 = GATLayer(in_dim, hidden)
        self.l2 = GATLayer(hidden, n_classes)

    def forward(self, x, adj):
        return self.l2(F.elu(self.l1(x, adj)), adj)


def gat_train_step(model, opt, x, adj, y):
    loss = F.cross_entropy(model(x, adj), y)
    loss.backward()
    opt.step()
    opt.zero_grad(True)
    return loss
