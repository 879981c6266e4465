#!/usr/bin/env python3
"""AUC-parity check (BASELINE.md metric 2): train the SAME Wide&Deep model
(a) through this framework's HIP bf16 kernel path and (b) as a stock
PyTorch fp32 eager implementation, on identical synthetic data with
identical initial weights, and compare validation AUC.

Usage: python tools/auc_parity.py [--rows 200000] [--epochs 3]
Prints one JSON line: {"auc_ours":..., "auc_stock":..., "delta":...}
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import numpy as np
import torch

from shifu_amd.data.csv_loader import TabularDataset
from shifu_amd.data.synthetic import synthetic_arrays
from shifu_amd.models.wide_deep import WideDeep
from shifu_amd.train.trainer import auc_score

N_DENSE, VOCAB, EMBED, TOWER = 40, (5000, 5000, 5000, 5000), 16, [128, 64]


class StockWideDeep(torch.nn.Module):
    """Plain nn.Linear / nn.Embedding twin; weights copied from ours."""

    def __init__(self, src: WideDeep):
        super().__init__()
        self.wide_cat = torch.nn.Embedding(src.wide_cat.total_rows, 1)
        self.wide_cat.weight.data.copy_(src.wide_cat.arena.data.float())
        self.emb = torch.nn.Embedding(src.embeddings.total_rows, EMBED)
        self.emb.weight.data.copy_(src.embeddings.arena.data.float())
        self.register_buffer("offsets", src.embeddings.offsets.clone())
        self.register_buffer("sizes", src.embeddings.sizes.clone())
        self.wide_dense = torch.nn.Linear(N_DENSE, 1)
        self.wide_dense.weight.data.copy_(src.wide_dense.weight.data.float())
        self.wide_dense.bias.data.copy_(src.wide_dense.bias.data.float())
        layers = []
        for l in src.tower:
            lin = torch.nn.Linear(l.in_features, l.out_features)
            lin.weight.data.copy_(l.weight.data.float())
            lin.bias.data.copy_(l.bias.data.float())
            layers += [lin, torch.nn.ReLU()]
        self.tower = torch.nn.Sequential(*layers)
        head = torch.nn.Linear(src.shifu_output_0.in_features, 1)
        head.weight.data.copy_(src.shifu_output_0.weight.data.float())
        head.bias.data.copy_(src.shifu_output_0.bias.data.float())
        self.head = head

    def forward(self, dense, cats):
        flat = cats.clamp(min=0) % self.sizes + self.offsets
        wide = self.wide_cat(flat).sum(dim=(1, 2)) + self.wide_dense(dense).reshape(-1)
        emb = self.emb(flat).reshape(dense.shape[0], -1)
        x = torch.cat([dense, emb], dim=1)
        return wide + self.head(self.tower(x)).reshape(-1)


def eval_auc(model, dense, cats, target, bs=65536):
    scores = []
    with torch.no_grad():
        for s in range(0, len(target), bs):
            logit = model(dense[s:s + bs], cats[s:s + bs]).float()
            scores.append(torch.sigmoid(logit).cpu().numpy())
    return auc_score(np.concatenate(scores), target.cpu().numpy())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=200_000)
    ap.add_argument("--epochs", type=int, default=3)
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--lr", type=float, default=1e-3)
    args = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    dev = "cuda" if on_gpu else "cpu"
    d, c, t, w = synthetic_arrays(args.rows, N_DENSE, VOCAB, seed=11, weighted=False)
    ds = TabularDataset(d, c, t, w)
    train, valid = ds.split(0.2, seed=1)

    torch.manual_seed(0)
    ours = WideDeep(N_DENSE, list(VOCAB), EMBED, TOWER, ["relu"] * len(TOWER),
                    seed=7)
    stock = StockWideDeep(ours).to(dev)
    ours = ours.to(dev)

    # ---- ours: framework path (bf16 HIP kernels on GPU) ----
    from shifu_amd.ops.flat import FlatParams, bind_mirrors, split_params
    from shifu_amd.ops.loss import weighted_loss
    from shifu_amd.ops.optim import FusedOptimizer
    if on_gpu:
        for p in ours.parameters():
            if getattr(p, "_is_embedding_arena", False):
                p.data = p.data.to(torch.bfloat16)
    dp, ep = split_params(ours)
    flat = FlatParams(dp, mirror_bf16=on_gpu)
    bind_mirrors(ours, flat)
    opt = FusedOptimizer(flat, ep, optimizer="adam", lr=args.lr, l2_reg=0.0,
                         emb_optimizer="adagrad", emb_lr=0.01)

    dtype = torch.bfloat16 if on_gpu else torch.float32
    trd = torch.from_numpy(train.dense).to(dev, dtype)
    trc = torch.from_numpy(train.cats).to(dev)
    trt = torch.from_numpy(train.target).to(dev)
    trw = torch.ones_like(trt)
    vad = torch.from_numpy(valid.dense).to(dev, dtype)
    vac = torch.from_numpy(valid.cats).to(dev)
    vat = torch.from_numpy(valid.target).to(dev)

    gen = np.random.default_rng(3)
    n = len(trt)
    for ep_i in range(args.epochs):
        perm = torch.from_numpy(gen.permutation(n)).to(dev)
        for s in range(0, n, args.batch):
            idx = perm[s:s + args.batch]
            loss = weighted_loss(ours(trd[idx], trc[idx]), trt[idx], trw[idx],
                                 "sigmoid_ce")
            loss.backward()
            flat.sync_grads()
            opt.step()
            opt.zero_grad()
    auc_ours = eval_auc(ours, vad, vac, vat)

    # ---- stock: fp32 eager + torch.optim ----
    trd32 = torch.from_numpy(train.dense).to(dev)
    vad32 = torch.from_numpy(valid.dense).to(dev)
    dense_p = [p for name, p in stock.named_parameters() if "cat" not in name
               and not name.startswith("emb")]
    sopt = torch.optim.Adam(dense_p, lr=args.lr)
    eopt = torch.optim.Adagrad([stock.emb.weight, stock.wide_cat.weight], lr=0.01)
    bce = torch.nn.BCEWithLogitsLoss()
    gen = np.random.default_rng(3)
    for ep_i in range(args.epochs):
        perm = torch.from_numpy(gen.permutation(n)).to(dev)
        for s in range(0, n, args.batch):
            idx = perm[s:s + args.batch]
            loss = bce(stock(trd32[idx], trc[idx]), trt[idx])
            loss.backward()
            sopt.step()
            eopt.step()
            sopt.zero_grad()
            eopt.zero_grad()
    auc_stock = eval_auc(stock, vad32, vac, vat)

    print(json.dumps({"metric": "auc_parity_wide_deep",
                      "auc_ours": auc_ours, "auc_stock": auc_stock,
                      "delta": auc_ours - auc_stock,
                      "rows": args.rows, "epochs": args.epochs,
                      "ours_dtype": str(dtype), "device": dev}), flush=True)


if __name__ == "__main__":
    main()
