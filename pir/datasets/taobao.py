"""Taobao-style CTR model with PIR-degraded click-history access.

Parity target: the reference's Taobao workload
(paper/experimental/batch_pir/modules/taobao_rec/taobao_rec_dataset_v2.py):
per-feature EmbeddingBag tables (user features, ad features, click
history); the click-history table is the large private one fetched
through batch PIR.
"""

import os
from typing import List, Optional

import numpy as np
import torch
import torch.nn as nn
from sklearn.metrics import roc_auc_score


class CTRModel(nn.Module):
    def __init__(self, field_sizes, hist_items, emsize=16, hidden=64):
        super().__init__()
        self.field_embs = nn.ModuleList(
            [nn.Embedding(sz, emsize) for sz in field_sizes])
        self.hist_bag = nn.EmbeddingBag(hist_items, emsize, mode="mean")
        self.mlp = nn.Sequential(
            nn.Linear(emsize * (len(field_sizes) + 1), hidden), nn.ReLU(),
            nn.Linear(hidden, 1),
        )

    def forward(self, fields, hist_flat, hist_off):
        embs = [emb(fields[:, i]) for i, emb in enumerate(self.field_embs)]
        embs.append(self.hist_bag(hist_flat, hist_off))
        return self.mlp(torch.cat(embs, dim=1)).squeeze(1)


def _synthetic_ctr(num_items, field_sizes, num_samples, hist_len, seed):
    rng = np.random.default_rng(seed)
    w = 1.0 / np.arange(1, num_items + 1)
    w /= w.sum()
    data = []
    for _ in range(num_samples):
        fields = [int(rng.integers(0, s)) for s in field_sizes]
        shift = int(rng.integers(0, num_items))
        hist = ((rng.choice(num_items, size=hist_len, p=w) + shift)
                % num_items).tolist()
        # click signal depends on a field and history cluster position
        p = 0.7 if (fields[0] % 2 == shift % 2) else 0.3
        data.append((fields, hist, float(rng.random() < p)))
    return data


class TaobaoDataset:
    def __init__(self, num_items=8192, field_sizes=(32, 16, 8),
                 num_samples=4000, hist_len=16,
                 data_path: Optional[str] = None, seed=0):
        self.num_items = num_items
        self.field_sizes = list(field_sizes)
        if data_path and os.path.exists(data_path):
            # a preprocessed trace would be loaded here; raw-Taobao CSV
            # parsing is deployment-specific
            raise NotImplementedError(
                "supply a preprocessed trace; raw-taobao CSV parsing is "
                "site-specific")
        samples = _synthetic_ctr(num_items, self.field_sizes, num_samples,
                                 hist_len, seed)
        split = int(len(samples) * 0.8)
        self.train_samples = samples[:split]
        self.eval_samples = samples[split:]
        self.model = CTRModel(self.field_sizes, num_items)
        self.num_entries = num_items

    @property
    def train_patterns(self) -> List[List[int]]:
        return [h for _, h, _ in self.train_samples]

    @property
    def eval_patterns(self) -> List[List[int]]:
        return [h for _, h, _ in self.eval_samples]

    def _batch(self, samples, degraded=None):
        fields = torch.tensor([f for f, _, _ in samples])
        flat, off, labels = [], [], []
        for i, (_, hist, lab) in enumerate(samples):
            keep = degraded[i] if degraded is not None else hist
            off.append(len(flat))
            flat.extend(keep if keep else [0])
            labels.append(lab)
        return (fields, torch.tensor(flat), torch.tensor(off),
                torch.tensor(labels))

    def train_model(self, epochs=2, lr=0.01, batch=256, checkpoint_dir=None):
        opt = torch.optim.Adam(self.model.parameters(), lr=lr)
        lossf = nn.BCEWithLogitsLoss()
        for epoch in range(epochs):
            for i in range(0, len(self.train_samples), batch):
                f, flat, off, lab = self._batch(self.train_samples[i:i + batch])
                opt.zero_grad()
                loss = lossf(self.model(f, flat, off), lab)
                loss.backward()
                opt.step()
            if checkpoint_dir:
                os.makedirs(checkpoint_dir, exist_ok=True)
                torch.save(self.model.state_dict(),
                           os.path.join(checkpoint_dir,
                                        "ctr_epoch=%d.pt" % epoch))

    def evaluate(self, optimizer=None):
        self.model.eval()
        degraded = None
        requested = recovered = 0
        if optimizer is not None:
            degraded = []
            for _, hist, _ in self.eval_samples:
                rec, s = optimizer.fetch(hist)
                requested += s["requested"]
                recovered += s["recovered"]
                degraded.append([h for h in hist if h in rec])
        f, flat, off, lab = self._batch(self.eval_samples, degraded)
        with torch.no_grad():
            scores = torch.sigmoid(self.model(f, flat, off)).numpy()
        try:
            auc = float(roc_auc_score(lab.numpy(), scores))
        except ValueError:
            auc = float("nan")
        res = {"metric": "auc", "value": auc}
        if optimizer is not None:
            res["recovery_rate"] = recovered / max(1, requested)
        return res


def initialize(**kw):
    return TaobaoDataset(**kw)
