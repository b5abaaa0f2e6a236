"""HTTP PIR server (deployment demo): one process per trust domain.

Each server holds the table and answers batched DPF queries; a client
that talks to two non-colluding deployments reconstructs entries by
subtracting the two share vectors.

Run (one per server, on different hosts in production):
    uvicorn gpudpf.server:build_app --factory --port 8100
    GPUDPF_TABLE=path.pt GPUDPF_PRF=AES128 uvicorn ...

API:
    GET  /info                     -> table shape, PRF, device
    POST /eval  {"keys": [[524 ints], ...]}  -> {"shares": [[e ints], ...]}

Client helper: `pir_fetch(urls, table_n, index, prf)` in this module.
"""

import base64
import os

import numpy as np
import torch

from gpudpf import DPF


def build_app(table=None, prf=None):
    from fastapi import FastAPI
    from pydantic import BaseModel

    if table is None:
        path = os.environ.get("GPUDPF_TABLE")
        if path:
            table = torch.load(path)
        else:  # demo table
            table = (torch.arange(65536 * 8, dtype=torch.int64) % (2**31)).to(
                torch.int32
            ).reshape(65536, 8)
    if prf is None:
        prf = getattr(DPF, "PRF_" + os.environ.get("GPUDPF_PRF", "AES128"))

    dpf = DPF(prf=prf)
    dpf.eval_init(table)
    use_gpu = torch.cuda.is_available()

    app = FastAPI(title="gpudpf PIR server")

    class EvalRequest(BaseModel):
        keys_b64: str  # base64 of int32[batch, 524] little-endian

    @app.get("/info")
    def info():
        return {
            "entries": dpf.table_num_entries,
            "entry_size": dpf.table_effective_entry_size,
            "prf": dpf.prf_method_string,
            "device": dpf.device if use_gpu else "cpu",
            "key_bytes": DPF.KEY_INTS * 4,
        }

    @app.post("/eval")
    def eval_keys(req: EvalRequest):
        raw = base64.b64decode(req.keys_b64)
        keys = torch.from_numpy(
            np.frombuffer(raw, dtype=np.int32).reshape(-1, DPF.KEY_INTS).copy()
        )
        shares = dpf.eval_gpu(keys) if use_gpu else dpf.eval_cpu(keys)
        return {
            "shares_b64": base64.b64encode(
                shares.to(torch.int32).numpy().tobytes()
            ).decode()
        }

    return app


def pir_fetch(urls, n, indices, prf=None, timeout=30.0):
    """Client: fetch table entries at `indices` from two PIR servers."""
    import httpx

    dpf = DPF(prf=prf)
    k1s, k2s = [], []
    for i in indices:
        k1, k2 = dpf.gen(i, n)
        k1s.append(k1)
        k2s.append(k2)

    def post(url, keys):
        blob = base64.b64encode(
            torch.stack(keys).numpy().astype(np.int32).tobytes()
        ).decode()
        r = httpx.post(url + "/eval", json={"keys_b64": blob}, timeout=timeout)
        r.raise_for_status()
        raw = base64.b64decode(r.json()["shares_b64"])
        return torch.from_numpy(
            np.frombuffer(raw, dtype=np.int32).reshape(len(keys), -1).copy()
        )

    a = post(urls[0], k1s)
    b = post(urls[1], k2s)
    return (a.to(torch.int64) - b.to(torch.int64)).to(torch.int32)
