"""Tensor-parallel sharding math, multi-process over gloo (CPU, world=2).

The GPU TP decode path uses exactly this arithmetic (engine._decode_step
with world>1): head-sharded QKV, row-parallel o/down with all-reduce,
vocab-sharded lm_head with all-gather.  Here the same sharding helpers
are exercised with torch CPU tensors so CI without a GPU covers the
distributed plumbing (SURVEY §4 distributed strategy)."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    import torch.distributed as dist

    from llm_np_cp_amd.parallel import tp

    r, w = tp.init_distributed(backend="gloo")
    assert (r, w) == (rank, world)

    rng = np.random.default_rng(0)  # same on all ranks
    H, I, V = 32, 64, 96
    x = rng.standard_normal(H).astype(np.float32)
    Wg = rng.standard_normal((I, H)).astype(np.float32)   # column-parallel
    Wd = rng.standard_normal((H, I)).astype(np.float32)   # row-parallel
    Wl = rng.standard_normal((V, H)).astype(np.float32)   # vocab-sharded

    # column-parallel: each rank computes a slice of the I dim
    g_loc = tp.shard_rows(Wg, rank, world) @ x            # (I/world,)
    # row-parallel: K sharded -> partial sums -> all-reduce
    d_part = tp.shard_cols(Wd, rank, world) @ g_loc
    t = torch.from_numpy(d_part.copy())
    tp.all_reduce(t)
    full_ref = Wd @ (Wg @ x)
    np.testing.assert_allclose(t.numpy(), full_ref, rtol=1e-4, atol=1e-4)

    # vocab shard + all-gather
    l_loc = torch.from_numpy((tp.shard_rows(Wl, rank, world) @ x).copy())
    out = torch.zeros(V)
    tp.all_gather_into(out, l_loc)
    np.testing.assert_allclose(out.numpy(), Wl @ x, rtol=1e-4, atol=1e-4)

    # broadcast: ranks agree on rank 0's token
    tok = torch.tensor([rank * 100 + 7], dtype=torch.int32)
    tp.broadcast(tok, 0)
    assert tok.item() == 7

    results[rank] = "ok"
    dist.destroy_process_group()


@pytest.mark.parametrize("world,port", [(2, 29531), (4, 29532)])
def test_tp_shard_math_gloo(world, port):
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_worker, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
        for r in range(world):
            assert results.get(r) == "ok", f"rank {r} failed"


def test_shard_roundtrip():
    from llm_np_cp_amd.parallel import tp

    w = np.arange(48, dtype=np.float32).reshape(8, 6)
    rows = [tp.shard_rows(w, r, 4) for r in range(4)]
    np.testing.assert_array_equal(np.concatenate(rows, 0), w)
    cols = [tp.shard_cols(w, r, 3) for r in range(3)]
    np.testing.assert_array_equal(np.concatenate(cols, 1), w)


def test_world1_collectives_are_noops():
    from llm_np_cp_amd.parallel import tp

    t = torch.ones(4)
    tp.all_reduce(t)
    np.testing.assert_array_equal(t.numpy(), np.ones(4))
    out = torch.zeros(4)
    tp.all_gather_into(out, torch.arange(4.0))
    np.testing.assert_array_equal(out.numpy(), np.arange(4.0))


def test_xgmi_fits_predicate():
    """Collective routing predicate (decode-sized -> one-shot, else
    RCCL): contiguity, 16-byte granularity, slot capacity, dtype."""
    from types import SimpleNamespace

    from llm_np_cp_amd.parallel.xgmi import XgmiComm

    d = SimpleNamespace(slot_bytes=1024)
    fits = XgmiComm.fits
    assert fits(d, torch.zeros(512, dtype=torch.bfloat16))   # == slot
    assert fits(d, torch.zeros(256, dtype=torch.float32))
    assert not fits(d, torch.zeros(600, dtype=torch.float32))   # > slot
    assert not fits(d, torch.zeros(520, dtype=torch.bfloat16))  # > slot
    assert not fits(d, torch.zeros(12, dtype=torch.bfloat16))   # % 16
    assert not fits(d, torch.zeros(8, dtype=torch.int32))       # dtype
    assert not fits(d, torch.zeros(8, 8,
                                   dtype=torch.bfloat16)[:, :4])  # strided
