"""MovieLens-style click recommender with PIR-degraded history access.

Parity target: the reference's ml-20m workload
(paper/experimental/batch_pir/modules/movielens_rec/movielens_dataset.py):
an EmbeddingBag tower over the user's recent item history + an MLP
predicts click (rating >= 4); history items the batch-PIR plan fails to
recover are dropped from the bag, and the AUC drop is the accuracy cost.
"""

import os
from typing import List, Optional

import numpy as np
import torch
import torch.nn as nn
from sklearn.metrics import roc_auc_score


class RecModel(nn.Module):
    def __init__(self, num_items, emsize=32, hidden=64):
        super().__init__()
        self.history_bag = nn.EmbeddingBag(num_items, emsize, mode="mean")
        self.target_emb = nn.Embedding(num_items, emsize)
        self.mlp = nn.Sequential(
            nn.Linear(2 * emsize, hidden), nn.ReLU(),
            nn.Linear(hidden, hidden), nn.ReLU(),
            nn.Linear(hidden, 1),
        )

    def forward(self, hist_flat, hist_offsets, target):
        h = self.history_bag(hist_flat, hist_offsets)
        t = self.target_emb(target)
        return self.mlp(torch.cat([h, t], dim=1)).squeeze(1)


def _synthetic_interactions(num_items, num_users, hist_len, seed):
    """Users with Zipf item tastes; click prob raised when target is near
    the user's taste cluster (gives AUC signal)."""
    rng = np.random.default_rng(seed)
    w = 1.0 / np.arange(1, num_items + 1)
    w /= w.sum()
    samples = []
    for u in range(num_users):
        shift = rng.integers(0, num_items)
        hist = (rng.choice(num_items, size=hist_len, p=w) + shift) % num_items
        target = int((rng.choice(num_items, p=w) + shift) % num_items)
        affinity = np.mean([min((h - target) % num_items,
                                (target - h) % num_items) for h in hist])
        p_click = 0.8 if affinity < num_items * 0.2 else 0.2
        label = float(rng.random() < p_click)
        samples.append((hist.tolist(), target, label))
    return samples


class MovieLensDataset:
    def __init__(self, num_items=4096, num_users=2000, hist_len=20,
                 data_path: Optional[str] = None, seed=0):
        self.num_items = num_items
        if data_path and os.path.exists(data_path):
            import pandas as pd

            df = pd.read_csv(data_path)  # userId,movieId,rating columns
            df["item"] = df["movieId"].astype("category").cat.codes
            self.num_items = int(df["item"].max()) + 1
            samples = []
            for _, grp in df.groupby("userId"):
                items = grp["item"].tolist()
                ratings = grp["rating"].tolist()
                if len(items) < hist_len + 1:
                    continue
                hist, target = items[:hist_len], items[hist_len]
                samples.append((hist, target, float(ratings[hist_len] >= 4)))
        else:
            samples = _synthetic_interactions(num_items, num_users, hist_len,
                                              seed)
        split = int(len(samples) * 0.8)
        self.train_samples = samples[:split]
        self.eval_samples = samples[split:]
        self.model = RecModel(self.num_items)
        self.num_entries = self.num_items

    @property
    def train_patterns(self) -> List[List[int]]:
        return [h for h, _, _ in self.train_samples]

    @property
    def eval_patterns(self) -> List[List[int]]:
        return [h for h, _, _ in self.eval_samples]

    def _batch(self, samples, degraded=None):
        flat, offsets, targets, labels = [], [], [], []
        for i, (hist, tgt, lab) in enumerate(samples):
            keep = degraded[i] if degraded is not None else hist
            offsets.append(len(flat))
            flat.extend(keep if keep else [0])
            targets.append(tgt)
            labels.append(lab)
        return (torch.tensor(flat), torch.tensor(offsets),
                torch.tensor(targets), torch.tensor(labels))

    def train_model(self, epochs=2, lr=0.01, batch=256, checkpoint_dir=None):
        opt = torch.optim.Adam(self.model.parameters(), lr=lr)
        lossf = nn.BCEWithLogitsLoss()
        for epoch in range(epochs):
            for i in range(0, len(self.train_samples), batch):
                chunk = self.train_samples[i : i + batch]
                flat, off, tgt, lab = self._batch(chunk)
                opt.zero_grad()
                out = self.model(flat, off, tgt)
                loss = lossf(out, lab)
                loss.backward()
                opt.step()
            if checkpoint_dir:
                os.makedirs(checkpoint_dir, exist_ok=True)
                torch.save(self.model.state_dict(),
                           os.path.join(checkpoint_dir,
                                        "recmodel_epoch=%d.pt" % epoch))

    def evaluate(self, optimizer=None):
        """AUC on held-out samples; unrecovered history items dropped."""
        self.model.eval()
        degraded = None
        requested = recovered = 0
        if optimizer is not None:
            degraded = []
            for hist, _, _ in self.eval_samples:
                rec, s = optimizer.fetch(hist)
                requested += s["requested"]
                recovered += s["recovered"]
                degraded.append([h for h in hist if h in rec])
        flat, off, tgt, lab = self._batch(self.eval_samples, degraded)
        with torch.no_grad():
            scores = torch.sigmoid(self.model(flat, off, tgt)).numpy()
        try:
            auc = float(roc_auc_score(lab.numpy(), scores))
        except ValueError:
            auc = float("nan")
        res = {"metric": "auc", "value": auc}
        if optimizer is not None:
            res["recovery_rate"] = recovered / max(1, requested)
        return res


def initialize(**kw):
    return MovieLensDataset(**kw)
