"""Word language model with PIR-degraded embedding access.

Parity target: the reference's WikiText-2 LSTM/Transformer workload
(paper/experimental/batch_pir/modules/language_model/*): the on-device LM
fetches token embeddings privately; tokens whose embeddings the batch-PIR
plan fails to recover are replaced by <unk> and the perplexity hit is the
accuracy cost.
"""

import math
import os
from typing import List, Optional

import torch
import torch.nn as nn

UNK_ID = 9  # parity with the reference's <unk> id convention


class RNNModel(nn.Module):
    """LSTM language model with optionally tied embeddings."""

    def __init__(self, vocab, emsize=200, nhid=200, nlayers=2, dropout=0.2,
                 tie_weights=True):
        super().__init__()
        self.encoder = nn.Embedding(vocab, emsize)
        self.rnn = nn.LSTM(emsize, nhid, nlayers, dropout=dropout,
                           batch_first=False)
        self.decoder = nn.Linear(nhid, vocab)
        self.drop = nn.Dropout(dropout)
        if tie_weights and nhid == emsize:
            self.decoder.weight = self.encoder.weight
        self.vocab = vocab

    def forward(self, x, hidden=None):
        emb = self.drop(self.encoder(x))
        out, hidden = self.rnn(emb, hidden)
        return self.decoder(self.drop(out)), hidden


class TransformerModel(nn.Module):
    def __init__(self, vocab, emsize=200, nhead=2, nhid=200, nlayers=2,
                 dropout=0.2):
        super().__init__()
        layer = nn.TransformerEncoderLayer(emsize, nhead, nhid, dropout)
        self.encoder = nn.Embedding(vocab, emsize)
        self.transformer = nn.TransformerEncoder(layer, nlayers)
        self.decoder = nn.Linear(emsize, vocab)
        self.vocab = vocab

    def forward(self, x, hidden=None):
        emb = self.encoder(x) * math.sqrt(self.encoder.embedding_dim)
        mask = nn.Transformer.generate_square_subsequent_mask(x.shape[0])
        return self.decoder(self.transformer(emb, mask)), None


class Vocab:
    """First-appearance word index (the reference's Dictionary,
    language_model/data.py:8-19)."""

    def __init__(self):
        self.word2idx = {}
        self.idx2word = []

    def add(self, word):
        if word not in self.word2idx:
            self.word2idx[word] = len(self.idx2word)
            self.idx2word.append(word)
        return self.word2idx[word]

    def __len__(self):
        return len(self.idx2word)


def tokenize_text(path, vocab=None):
    """Whitespace-tokenize a text file, appending <eos> per line (the
    reference's Corpus.tokenize, language_model/data.py:28-49): returns
    (int64 token tensor, vocab).  WikiText-2's train.txt/valid.txt are in
    exactly this format."""
    if vocab is None:
        vocab = Vocab()
    ids = []
    with open(path, "r", encoding="utf8") as f:
        for line in f:
            for word in line.split() + ["<eos>"]:
                ids.append(vocab.add(word))
    return torch.tensor(ids, dtype=torch.long), vocab


def load_text_corpus(data_path):
    """Real-text corpus: data_path may be a WikiText-2-style directory
    (train.txt + valid.txt [+ test.txt], one shared vocab) or a single
    .txt file (90/10 split).  Returns (train_tokens, val_tokens, vocab)."""
    if os.path.isdir(data_path):
        vocab = Vocab()
        train, vocab = tokenize_text(os.path.join(data_path, "train.txt"),
                                     vocab)
        val, vocab = tokenize_text(os.path.join(data_path, "valid.txt"),
                                   vocab)
        return train, val, vocab
    tokens, vocab = tokenize_text(data_path)
    split = int(tokens.shape[0] * 0.9)
    return tokens[:split], tokens[split:], vocab


def _synthetic_corpus(vocab, length, seed):
    """Zipf-distributed token stream (same shape as a tokenized corpus)."""
    g = torch.Generator().manual_seed(seed)
    w = 1.0 / torch.arange(1, vocab + 1, dtype=torch.float64)
    return torch.multinomial(w, length, replacement=True, generator=g).to(
        torch.long
    )


def batchify(data, bsz):
    nbatch = data.shape[0] // bsz
    return data[: nbatch * bsz].view(bsz, -1).t().contiguous()


class LanguageModelDataset:
    def __init__(self, vocab=2048, bptt=35, batch_size=20, model="lstm",
                 data_path: Optional[str] = None, seed=0, corpus_len=200000):
        self.vocab = vocab
        self.bptt = bptt
        self.batch_size = batch_size
        self.vocab_words = None
        if data_path and os.path.exists(data_path):
            if data_path.endswith(".pt"):
                tokens = torch.load(data_path)
                self.vocab = int(tokens.max().item()) + 1
                split = int(tokens.shape[0] * 0.9)
                train_tokens, val_tokens = tokens[:split], tokens[split:]
            else:
                # real text: WikiText-2-style directory or a .txt file
                train_tokens, val_tokens, v = load_text_corpus(data_path)
                self.vocab = len(v)
                self.vocab_words = v
        else:
            tokens = _synthetic_corpus(vocab, corpus_len, seed)
            split = int(tokens.shape[0] * 0.9)
            train_tokens, val_tokens = tokens[:split], tokens[split:]
        self.train_data = batchify(train_tokens, batch_size)
        self.val_data = batchify(val_tokens, batch_size)
        self.unk_id = min(UNK_ID, self.vocab - 1)
        if self.vocab_words is not None and "<unk>" in self.vocab_words.word2idx:
            self.unk_id = self.vocab_words.word2idx["<unk>"]
        cls = RNNModel if model == "lstm" else TransformerModel
        self.model = cls(self.vocab)
        self.num_entries = self.vocab
        self.criterion = nn.CrossEntropyLoss()

    # -- access patterns: one inference = one bptt window of token ids ----
    def _patterns(self, data) -> List[List[int]]:
        pats = []
        for i in range(0, data.shape[0] - 1, self.bptt):
            chunk = data[i : i + self.bptt]
            for col in range(chunk.shape[1]):
                pats.append(chunk[:, col].tolist())
        return pats

    @property
    def train_patterns(self):
        return self._patterns(self.train_data)

    @property
    def eval_patterns(self):
        return self._patterns(self.val_data)

    # -- training ----------------------------------------------------------
    def train_model(self, epochs=1, lr=5.0, max_batches=None,
                    checkpoint_dir=None):
        model = self.model
        model.train()
        for epoch in range(epochs):
            total, count = 0.0, 0
            hidden = None
            for bi, i in enumerate(range(0, self.train_data.shape[0] - 1,
                                         self.bptt)):
                if max_batches and bi >= max_batches:
                    break
                x = self.train_data[i : i + self.bptt]
                y = self.train_data[i + 1 : i + 1 + self.bptt]
                if y.shape[0] != x.shape[0]:
                    break
                model.zero_grad()
                out, hidden = model(x, None)
                loss = self.criterion(out.view(-1, self.vocab), y.reshape(-1))
                loss.backward()
                torch.nn.utils.clip_grad_norm_(model.parameters(), 0.25)
                with torch.no_grad():
                    for p in model.parameters():
                        if p.grad is not None:
                            p.add_(p.grad, alpha=-lr / 100)
                total += loss.item()
                count += 1
            if checkpoint_dir:
                os.makedirs(checkpoint_dir, exist_ok=True)
                torch.save(model.state_dict(),
                           os.path.join(checkpoint_dir,
                                        "lm_epoch=%d.pt" % epoch))
        return total / max(1, count)

    # -- PIR-degraded evaluation -------------------------------------------
    def evaluate(self, optimizer=None, max_batches=50):
        """Perplexity on val data; when a BatchPIROptimize plan is given,
        input tokens whose embeddings are unrecovered become <unk>."""
        model = self.model
        model.eval()
        total, count = 0.0, 0
        requested = recovered = 0
        with torch.no_grad():
            for bi, i in enumerate(range(0, self.val_data.shape[0] - 1,
                                         self.bptt)):
                if bi >= max_batches:
                    break
                x = self.val_data[i : i + self.bptt].clone()
                y = self.val_data[i + 1 : i + 1 + self.bptt]
                if y.shape[0] != x.shape[0]:
                    break
                if optimizer is not None:
                    for col in range(x.shape[1]):
                        idxs = x[:, col].tolist()
                        rec, s = optimizer.fetch(idxs)
                        requested += s["requested"]
                        recovered += s["recovered"]
                        for r in range(x.shape[0]):
                            if int(x[r, col]) not in rec:
                                x[r, col] = self.unk_id
                out, _ = model(x, None)
                loss = self.criterion(out.view(-1, self.vocab), y.reshape(-1))
                total += loss.item()
                count += 1
        ppl = math.exp(min(20.0, total / max(1, count)))
        res = {"metric": "ppl", "value": ppl}
        if optimizer is not None:
            res["recovery_rate"] = recovered / max(1, requested)
        return res


def initialize(**kw):
    return LanguageModelDataset(**kw)
