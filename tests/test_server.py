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


@pytest.mark.gpu
def test_pir_http_roundtrip_gpu():
    """Same HTTP layer exercising the GPU eval path (build_app routes to
    eval_gpu when a GPU is present)."""
    assert torch.cuda.is_available()
    n, e = 4096, 16
    torch.manual_seed(13)
    table = torch.randint(-(2**31), 2**31 - 1, (n, e), dtype=torch.int64).to(
        torch.int32)
    app = build_app(table=table, prf=DPF.PRF_AES128)
    client = TestClient(app)
    assert client.get("/info").json()["device"].startswith("cuda")

    dpf = DPF(prf=DPF.PRF_AES128)
    idxs = [0, 4095, 1234]
    pairs = [dpf.gen(i, n) for i in idxs]

    def post(keys):
        blob = base64.b64encode(
            torch.stack(keys).numpy().astype(np.int32).tobytes()).decode()
        r = client.post("/eval", json={"keys_b64": blob})
        assert r.status_code == 200
        raw = base64.b64decode(r.json()["shares_b64"])
        return torch.from_numpy(
            np.frombuffer(raw, dtype=np.int32).reshape(len(keys), -1).copy())

    a = post([p[0] for p in pairs])
    b = post([p[1] for p in pairs])
    rec = (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32)
    assert torch.equal(rec, table[idxs, :])


def test_pir_fetch_client_helper(monkeypatch):
    """The 2-server client helper end to end: pir_fetch's HTTP calls are
    routed to two in-process TestClient apps (distinct trust domains)."""
    import gpudpf.server as srv_mod

    n, e = 1024, 8
    torch.manual_seed(3)
    table = torch.randint(-(2**31), 2**31 - 1, (n, e), dtype=torch.int64).to(
        torch.int32)
    apps = {
        "http://server-a": TestClient(build_app(table=table,
                                                prf=DPF.PRF_SALSA20)),
        "http://server-b": TestClient(build_app(table=table,
                                                prf=DPF.PRF_SALSA20)),
    }

    class _FakeHttpx:
        @staticmethod
        def post(url, json=None, timeout=None):
            base, path = url.rsplit("/", 1)
            return apps[base].post("/" + path, json=json)

    monkeypatch.setitem(__import__("sys").modules, "httpx", _FakeHttpx)
    idxs = [0, 511, 1023]
    got = srv_mod.pir_fetch(["http://server-a", "http://server-b"], n, idxs,
                            prf=DPF.PRF_SALSA20)
    assert torch.equal(got, table[idxs, :])
