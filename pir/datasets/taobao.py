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


def _remap_columns(rows):
    """Per-column categorical remap to dense first-appearance indices
    (the reference's process_features_list semantics,
    taobao_rec_dataset_v2.py:67-85)."""
    cols = len(rows[0])
    maps = [dict() for _ in range(cols)]
    for r in rows:
        for c in range(cols):
            if r[c] not in maps[c]:
                maps[c][r[c]] = len(maps[c])
    remapped = [[maps[c][r[c]] for c in range(cols)] for r in rows]
    return remapped, maps


def load_taobao_csvs(path, sample_limit=None, hist_len=16):
    """Real Taobao CTR pipeline (reference:
    taobao_rec_dataset_v2.py:87-197): reads the three CSVs of the Taobao
    display-ad dataset, remaps every categorical column to dense indices,
    joins click events with user/ad profiles (skipping events whose
    profile rows are missing), and accumulates each user's click-history
    access pattern over the remapped ad ids.

    ad_feature.csv:   adgroup_id,cate_id,campaign_id,customer,brand,price
                      (brand NULL -> 0; the dense price column is dropped
                      from the sparse feature set)
    user_profile.csv: userid + 8 categorical features (empty -> 0)
    raw_sample.csv:   user,time_stamp,adgroup_id,pid,nonclk,clk

    Returns (samples, field_sizes, num_items): samples are
    (fields, history_before_event, clk) with history capped at hist_len.
    """
    def rows_of(name, parse):
        out = []
        with open(os.path.join(path, name), "r", encoding="utf8") as f:
            for i, line in enumerate(f):
                if i == 0:
                    continue  # header
                parts = line.rstrip("\n").split(",")
                out.append(parse(parts))
        return out

    def ad_parse(p):
        brand = 0 if p[4].strip() in ("", "NULL") else int(float(p[4]))
        return [int(p[0]), int(p[1]), int(p[2]), int(p[3]), brand]

    def user_parse(p):
        return [0 if x.strip() in ("", "NULL") else int(float(x)) for x in p]

    ads_raw = rows_of("ad_feature.csv", ad_parse)
    users_raw = rows_of("user_profile.csv", user_parse)
    ads, ad_maps = _remap_columns(ads_raw)
    users, user_maps = _remap_columns(users_raw)
    ad_by_id = {r[0]: r for r in ads}
    user_by_id = {r[0]: r for r in users}

    samples = []
    histories = {}
    events = rows_of(
        "raw_sample.csv",
        lambda p: (int(p[0]), int(p[1]), int(p[2]), int(p[5])))
    events.sort(key=lambda ev: ev[1])  # chronological
    for uid_raw, _ts, aid_raw, clk in events:
        if sample_limit is not None and len(samples) >= sample_limit:
            break
        if uid_raw not in user_maps[0] or aid_raw not in ad_maps[0]:
            continue  # profile row missing: skip, as the reference does
        uid = user_maps[0][uid_raw]
        aid = ad_maps[0][aid_raw]
        fields = ad_by_id[aid] + user_by_id[uid]
        hist = histories.setdefault(uid, [])
        samples.append((fields, list(hist[-hist_len:]), float(clk)))
        hist.append(aid)  # history grows AFTER the event (no label leak)

    field_sizes = [len(m) for m in ad_maps] + [len(m) for m in user_maps]
    num_items = len(ad_maps[0])
    return samples, field_sizes, num_items


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
            samples, self.field_sizes, self.num_items = load_taobao_csvs(
                data_path, sample_limit=num_samples, hist_len=hist_len)
            num_items = self.num_items
        else:
            samples = _synthetic_ctr(num_items, self.field_sizes,
                                     num_samples, hist_len, seed)
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
