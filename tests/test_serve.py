"""Serving stack: dynamic micro-batching engine + HTTP app + TP fan-out.

Beyond-reference subsystem (pipegoose_amd/serve.py) — the reference has
no serving path; these tests pin the batching engine to the direct
``generate()`` oracle (greedy decode is deterministic) and the TP wire
protocol to the collective call.
"""
import threading

import pytest
import torch

from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
from pipegoose_amd.serve import BatchingEngine, make_app, worker_loop
from pipegoose_amd.testing.utils import init_parallel_context, spawn


def _tiny_model(ctx):
    torch.manual_seed(0)
    return BloomForCausalLM(bloom_tiny(), ctx).eval()


def _ctx1(rank=0, world_size=1, port=29511):
    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(port))
    return init_parallel_context(rank, world_size, port)


def test_engine_batches_and_matches_direct():
    ctx = _ctx1()
    try:
        model = _tiny_model(ctx)
        engine = BatchingEngine(model, parallel_context=ctx, max_batch=4,
                                max_wait_ms=300.0)  # generous: slow CI boxes
                                                    # must still coalesce
        torch.manual_seed(1)
        prompts = [torch.randint(0, 100, (8,)) for _ in range(3)]
        odd = torch.randint(0, 100, (5,))          # different length

        with torch.no_grad():
            want = model.generate(torch.stack(prompts), max_new_tokens=6)
            want_odd = model.generate(odd.unsqueeze(0), max_new_tokens=6)

        results = {}

        def worker(i, ids):
            results[i] = engine.submit(ids, max_new_tokens=6)

        threads = [threading.Thread(target=worker, args=(i, p))
                   for i, p in enumerate(prompts)]
        threads.append(threading.Thread(target=worker, args=(3, odd)))
        for t in threads:
            t.start()
        for t in threads:
            t.join(60)

        for i in range(3):
            assert torch.equal(results[i], want[i]), f"request {i} mismatch"
        assert torch.equal(results[3], want_odd[0])
        # the three same-shape requests must have been coalesced
        assert engine.n_requests == 4
        assert engine.n_batches <= 3
        engine.shutdown()
    finally:
        ctx.destroy()


def test_engine_surfaces_errors():
    ctx = _ctx1(port=29512)
    try:
        model = _tiny_model(ctx)
        engine = BatchingEngine(model, parallel_context=ctx)
        with pytest.raises(RuntimeError):
            # vocab overflow -> embedding index error, surfaced not fatal
            engine.submit(torch.tensor([10 ** 6]), max_new_tokens=2)
        # engine still alive afterwards
        out = engine.submit(torch.tensor([1, 2, 3]), max_new_tokens=2)
        assert out.numel() == 5
        engine.shutdown()
    finally:
        ctx.destroy()


def test_http_app():
    fastapi = pytest.importorskip("fastapi")  # noqa: F841
    from fastapi.testclient import TestClient

    ctx = _ctx1(port=29513)
    try:
        model = _tiny_model(ctx)
        engine = BatchingEngine(model, parallel_context=ctx)
        app = make_app(engine)
        client = TestClient(app)

        r = client.get("/healthz")
        assert r.status_code == 200 and r.json()["status"] == "ok"

        rows = [[1, 2, 3, 4], [5, 6, 7, 8]]
        r = client.post("/generate",
                        json={"input_ids": rows, "max_new_tokens": 4})
        assert r.status_code == 200
        out = r.json()["output_ids"]
        assert len(out) == 2 and all(len(o) == 8 for o in out)
        with torch.no_grad():
            want = model.generate(torch.tensor(rows), max_new_tokens=4)
        assert out == want.tolist()

        r = client.post("/generate", json={})
        assert r.status_code == 400
        engine.shutdown()
    finally:
        ctx.destroy()


def _tp2_serve(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port,
                                tensor_parallel_size=2)
    torch.manual_seed(0)
    # the native family is TP-sharded by construction when ctx has tp>1
    model = BloomForCausalLM(bloom_tiny(), ctx).eval()
    torch.manual_seed(2)
    ids = torch.randint(0, 100, (1, 8))

    # oracle: all ranks run the collective generate directly
    with torch.no_grad():
        want = model.generate(ids, max_new_tokens=5)

    if rank == 0:
        engine = BatchingEngine(model, parallel_context=ctx)
        got = engine.submit(ids[0], max_new_tokens=5)
        assert torch.equal(got, want[0]), "TP engine result != collective"
        engine.shutdown()        # broadcasts the stop header to rank 1
    else:
        worker_loop(model, ctx)  # returns on the stop header
    ctx.destroy()


def test_tp2_engine_fanout():
    spawn(_tp2_serve, world_size=2)
