#!/usr/bin/env python3
"""Grid-barrier cost microbenchmark (gates the fused-layer decode
design: a phase boundary must beat the ~4 us kernel launch floor)."""
import ctypes
import sys
import time

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from csrc.build import ensure_built  # noqa: E402

ensure_built()
from llm_np_cp_amd.ops import hip_ops as ho  # noqa: E402

lib = ho.lib()
lib.launch_gbar_bench.argtypes = [ctypes.c_void_p] * 3 + \
    [ctypes.c_int, ctypes.c_int, ctypes.c_long, ctypes.c_void_p]
lib.launch_gbar_bench.restype = ctypes.c_int

dev = torch.device("cuda:0")
cnt = torch.zeros(1, dtype=torch.int64, device=dev)
seq = torch.zeros(1, dtype=torch.int64, device=dev)
err = torch.zeros(1, dtype=torch.int32, device=dev)

for nblocks in (256, 512, 768, 1024):
    for nbar in (64,):
        cnt.zero_(); seq.zero_(); err.zero_()
        s = ho._stream()
        # warmup
        lib.launch_gbar_bench(cnt.data_ptr(), seq.data_ptr(),
                              err.data_ptr(), nblocks, nbar,
                              50_000_000, s)
        torch.cuda.synchronize()
        assert err.item() == 0, f"warmup barrier timed out at {nblocks}"
        reps = 20
        t0 = time.perf_counter()
        for _ in range(reps):
            lib.launch_gbar_bench(cnt.data_ptr(), seq.data_ptr(),
                                  err.data_ptr(), nblocks, nbar,
                                  50_000_000, s)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        assert err.item() == 0, f"barrier timed out at {nblocks}"
        per_kernel = dt / reps * 1e6
        per_bar = (per_kernel) / nbar
        print(f"nblocks={nblocks:5d} nbar={nbar}: {per_kernel:8.1f} us/kernel"
              f" -> {per_bar:6.2f} us/barrier (incl launch amortized)")
