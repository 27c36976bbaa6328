"""One-shot xGMI collective tests (1 GPU).

The kernels are validated three ways without needing an 8-GPU node:
  1. eight SIMULATED ranks in one process (8 comm buffers + 8 streams on
     one GPU) run the real kernel, ticket protocol, epoch parity and
     graph replay;
  2. two PROCESSES on one GPU exchange real hipIpc (dmabuf) handles and
     run XgmiComm.create()'s hardware self-check against gloo;
  3. a full TP=2 engine decode (two processes, one GPU) captures the
     one-shot collectives inside a hipGraph and replays it — the
     VERDICT r1 'graph capture x collectives' unknown, resolved on
     hardware.
Cross-DEVICE coherence still needs a multi-GPU node (driver round-end);
XgmiComm.create() re-runs the same self-check there at engine init.
"""

import os
import struct

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _ho():
    from llm_np_cp_amd.ops import hip_ops as ho
    return ho


# ONE shared stream pool for every simulated-rank test: spinning kernels
# must be co-resident, and HIP maps streams round-robin onto a bounded
# set of hardware queues (GPU_MAX_HW_QUEUES, raised in conftest.py) — a
# fresh 8 streams per test would eventually collide two spinners onto
# one queue and serialize them into a timeout.
_STREAMS = []


def _streams(world):
    while len(_STREAMS) < world:
        _STREAMS.append(torch.cuda.Stream(device=torch.device("cuda:0")))
    return _STREAMS[:world]


def _sim_world(ho, world, slot_bytes):
    total = ho.XC_OFF_DATA + 2 * 8 * slot_bytes
    bases = [ho.xc_alloc(total) for _ in range(world)]
    table = struct.pack("<8Q", *(bases + [0] * (8 - world)))
    for b in bases:
        ho.xc_memset(b, 0, ho.XC_OFF_DATA)
        ho.xc_h2d(b, table)
    return bases


def _free_world(ho, bases):
    for b in bases:
        ho.xc_free(b)


@pytest.mark.parametrize("world,n,nstripes", [(8, 3584, 1), (8, 4096, 2),
                                              (2, 2048, 1), (4, 8192, 4)])
def test_oneshot_allreduce_bf16_simulated(world, n, nstripes):
    ho = _ho()
    slot = 1 << 16
    bases = _sim_world(ho, world, slot)
    try:
        dev = torch.device("cuda:0")
        g = torch.Generator(device="cpu").manual_seed(7)
        srcs = [torch.randn(n, generator=g).to(dev).to(torch.bfloat16)
                for _ in range(world)]
        expect = sum(s.float() for s in srcs)
        bufs = [s.clone() for s in srcs]
        streams = _streams(world)
        for r in range(world):
            with torch.cuda.stream(streams[r]):
                ho.xgmi_coll(bufs[r].data_ptr(), bufs[r].data_ptr(),
                             bases[r], r, world, n * 2, slot, 0, nstripes,
                             spin_limit=20_000_000)
        torch.cuda.synchronize()
        for r in range(world):
            err = struct.unpack("<I",
                                ho.xc_d2h(bases[r] + ho.XC_OFF_ERR, 4))[0]
            assert err == 0, f"rank {r} one-shot timed out"
            assert torch.allclose(bufs[r].float(), expect, atol=3e-1,
                                  rtol=2e-2), f"rank {r} mismatch"
    finally:
        _free_world(ho, bases)


def test_oneshot_f32_and_gather_simulated():
    ho = _ho()
    world, n = 8, 4096
    slot = 1 << 16
    bases = _sim_world(ho, world, slot)
    try:
        dev = torch.device("cuda:0")
        g = torch.Generator(device="cpu").manual_seed(9)
        srcs = [torch.randn(n, generator=g).to(dev) for _ in range(world)]
        expect = sum(srcs)
        outs = [s.clone() for s in srcs]
        gouts = [torch.zeros(world * n, device=dev) for _ in range(world)]
        streams = _streams(world)
        # f32 all-reduce then gather back-to-back on each stream:
        # exercises epoch parity alternation within one submission
        for r in range(world):
            with torch.cuda.stream(streams[r]):
                ho.xgmi_coll(outs[r].data_ptr(), outs[r].data_ptr(),
                             bases[r], r, world, n * 4, slot, 1, 2,
                             spin_limit=20_000_000)
                ho.xgmi_coll(gouts[r].data_ptr(), srcs[r].data_ptr(),
                             bases[r], r, world, n * 4, slot, 2, 2,
                             spin_limit=20_000_000)
        torch.cuda.synchronize()
        cat = torch.cat(srcs)
        for r in range(world):
            err = struct.unpack("<I",
                                ho.xc_d2h(bases[r] + ho.XC_OFF_ERR, 4))[0]
            assert err == 0
            # fp32 sum, fixed order -> exact and identical across ranks
            assert torch.equal(outs[r], outs[0])
            assert torch.allclose(outs[r], expect, atol=1e-4, rtol=1e-5)
            assert torch.equal(gouts[r], cat)
    finally:
        _free_world(ho, bases)


def test_oneshot_graph_replay_simulated():
    """Capture the one-shot all-reduce into per-rank hipGraphs and
    replay 3x: epoch/ticket re-arm must keep results correct (the graph
    path the TP decode step uses)."""
    ho = _ho()
    world, n = 8, 3584
    slot = 1 << 16
    bases = _sim_world(ho, world, slot)
    try:
        dev = torch.device("cuda:0")
        gen = torch.Generator(device="cpu").manual_seed(11)
        srcs = [torch.randn(n, generator=gen).to(dev).to(torch.bfloat16)
                for _ in range(world)]
        bufs = [torch.empty_like(s) for s in srcs]
        expect = sum(s.float() for s in srcs)
        streams = _streams(world)

        def _coll(r):
            ho.xgmi_coll(bufs[r].data_ptr(), bufs[r].data_ptr(), bases[r],
                         r, world, n * 2, slot, 0, 2,
                         spin_limit=20_000_000)

        # eager warmup round (epoch 1)
        for r in range(world):
            bufs[r].copy_(srcs[r])
            with torch.cuda.stream(streams[r]):
                _coll(r)
        torch.cuda.synchronize()
        graphs = []
        for r in range(world):
            bufs[r].copy_(srcs[r])
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, stream=streams[r]):
                _coll(r)
            graphs.append(g)
        for rep in range(3):
            for r in range(world):
                bufs[r].copy_(srcs[r])
            torch.cuda.synchronize()
            for r in range(world):
                with torch.cuda.stream(streams[r]):
                    graphs[r].replay()
            torch.cuda.synchronize()
            for r in range(world):
                err = struct.unpack(
                    "<I", ho.xc_d2h(bases[r] + ho.XC_OFF_ERR, 4))[0]
                assert err == 0, f"replay {rep} rank {r} timed out"
                assert torch.allclose(bufs[r].float(), expect, atol=3e-1,
                                      rtol=2e-2), f"replay {rep} rank {r}"
        # epochs advanced: warmup + capture(0) + 3 replays = 4 per rank
        for r in range(world):
            ep = struct.unpack("<Q",
                               ho.xc_d2h(bases[r] + ho.XC_OFF_EPOCH, 8))[0]
            assert ep == 4, f"rank {r} epoch {ep} != 4"
    finally:
        _free_world(ho, bases)


# ----------------------------------------------------------------------
# two processes, one GPU: real hipIpc (dmabuf) handle exchange
# ----------------------------------------------------------------------

def _ipc_worker(rank, world, port, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    try:
        import torch
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)
        from llm_np_cp_amd.parallel.xgmi import XgmiComm
        comm = XgmiComm.create(rank, world, torch.device("cuda:0"),
                               slot_bytes=1 << 16)
        assert comm is not None, "XgmiComm.create failed (IPC or numerics)"
        # extra round on top of create()'s self-check: distinct values
        t = torch.full((1024,), float(rank + 1), device="cuda:0",
                       dtype=torch.bfloat16)
        comm.all_reduce(t, spin_limit=50_000_000)
        torch.cuda.synchronize()
        comm.check()
        expect = sum(range(1, world + 1))
        assert torch.all(t.float() == expect), t[:4]
        dist.barrier()
        comm.close()
        q.put((rank, "ok"))
    except Exception as e:  # surface the real error in the parent
        import traceback
        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def test_xgmi_comm_two_processes_one_gpu():
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    world = 2
    procs = [ctx.Process(target=_ipc_worker, args=(r, world, 29631, q))
             for r in range(world)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(world):
        r, msg = q.get()
        outs[r] = msg
    for p in procs:
        p.join(timeout=120)
    assert all(m == "ok" for m in outs.values()), outs


# ----------------------------------------------------------------------
# full TP=2 engine on one GPU: graph-captured decode with one-shot
# collectives inside the graph
# ----------------------------------------------------------------------

def _tp2_worker(rank, world, port, q):  # noqa: C901
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    try:
        import numpy as np
        import torch
        import torch.distributed as dist
        dist.init_process_group("gloo", rank=rank, world_size=world)
        torch.cuda.set_device(0)
        import llm_np_cp_amd as L
        from llm_np_cp_amd.io.loader import random_weights
        from llm_np_cp_amd.models.engine import GPUModel
        from llm_np_cp_amd.parallel import tp

        preset = "tiny-llama-tp" if world <= 2 else "tiny-llama-tp4"
        cfg = L.preset_config(preset)
        w = random_weights(cfg, seed=0)
        model = GPUModel(cfg, w, max_seq=128, device="cuda:0")
        assert tp.xgmi_comm() is not None, \
            "one-shot comm must come up for TP on-GPU decode"
        prompt = np.arange(1, 9, dtype=np.int32)
        cache, logits = model.prefill(prompt)
        ids = model.decode(6, greedy=True, use_graph=True)
        assert model._graph is not None and model._graph_mode is not None, \
            "decode must run from a captured graph (no eager fallback)"
        assert not getattr(model, "_graph_failed", False), \
            f"graph capture failed: {getattr(model, '_graph_error', None)}"
        dist.barrier()
        q.put((rank, "ok", logits.ravel().astype(np.float32),
               np.asarray(ids, dtype=np.int64)))
    except Exception as e:
        import traceback
        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}", None, None))


@pytest.mark.parametrize("world", [2, 4])
def test_tp_engine_graph_decode_one_gpu(world):
    import llm_np_cp_amd as L
    from llm_np_cp_amd.io.loader import random_weights
    from llm_np_cp_amd.models.engine import GPUModel

    # TP=1 reference in the parent (same weights)
    cfg = L.preset_config("tiny-llama-tp" if world <= 2
                          else "tiny-llama-tp4")
    w = random_weights(cfg, seed=0)
    ref = GPUModel(cfg, w, max_seq=128, device="cuda:0")
    prompt = np.arange(1, 9, dtype=np.int32)
    _, ref_logits = ref.prefill(prompt)
    ref_logits = ref_logits.ravel().astype(np.float32)
    del ref
    torch.cuda.empty_cache()

    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_tp2_worker,
                         args=(r, world, 29637 + world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    outs = {}
    for _ in range(world):
        item = q.get()
        outs[item[0]] = item[1:]
    for p in procs:
        p.join(timeout=300)
    for r, item in outs.items():
        assert item[0] == "ok", f"rank {r}: {item[0]}"
    l0, ids0 = outs[0][1], outs[0][2]
    for r in range(1, world):
        # TP ranks are bitwise-identical (fixed-order one-shot reduction)
        assert np.array_equal(ids0, outs[r][2])
        assert np.array_equal(l0, outs[r][1])
    # TP=2 vs TP=1 logits agree within bf16 partial-sum reordering noise
    scale = np.abs(ref_logits).max() + 1e-6
    assert np.abs(l0 - ref_logits).max() < 0.05 * scale + 0.05, \
        np.abs(l0 - ref_logits).max()
