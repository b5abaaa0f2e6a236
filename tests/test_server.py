"""HTTP PIR server tests (in-process, CPU path)."""

import base64

import numpy as np
import torch
import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from gpudpf import DPF  # noqa: E402
from gpudpf.server import build_app  # noqa: E402


def test_pir_http_roundtrip():
    n, e = 512, 4
    table = torch.arange(n * e, dtype=torch.int32).reshape(n, e)
    app = build_app(table=table, prf=DPF.PRF_SALSA20)
    client = TestClient(app)

    info = client.get("/info").json()
    assert info["entries"] == n and info["entry_size"] == e
    assert info["key_bytes"] == 2096

    dpf = DPF(prf=DPF.PRF_SALSA20)
    idxs = [3, 511, 100]
    k1s, k2s = [], []
    for i in idxs:
        k1, k2 = dpf.gen(i, n)
        k1s.append(k1)
        k2s.append(k2)

    def post(keys):
        blob = base64.b64encode(
            torch.stack(keys).numpy().astype(np.int32).tobytes()
        ).decode()
        r = client.post("/eval", json={"keys_b64": blob})
        assert r.status_code == 200
        raw = base64.b64decode(r.json()["shares_b64"])
        return torch.from_numpy(
            np.frombuffer(raw, dtype=np.int32).reshape(len(keys), -1).copy()
        )

    a = post(k1s)
    b = post(k2s)
    rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[idxs, :])
