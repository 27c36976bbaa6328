// xgmi_comm.hip — one-shot peer-memory collectives for single-node TP
// over xGMI (MI355X: 7 point-to-point links/GPU, ~153 GB/s each).
//
// Why not RCCL in the decode hot loop:
//  - decode TP payloads are hidden_size bf16 = 4-9 KB -> latency-bound;
//    a ring all-reduce pays 2(world-1) link hops, ~10-20 us per call,
//    and a TP=8 Gemma-9B step issues 84 of them.  One-shot over
//    peer-mapped HBM (every rank pushes its full vector to all peers,
//    every rank reduces locally) is one xGMI hop, a few us.
//  - these kernels are plain HIP launches on the current stream, so the
//    whole TP decode step captures into a hipGraph — no RCCL-in-graph
//    unknown.  Epoch/ticket state lives in device memory and the kernel
//    re-arms it, so graph REPLAY is safe (same recipe as k_attn_dec's
//    G16 ticket).
//
// Memory/coherence protocol (release/acquire over the HIP memory model):
//  writer: plain 16-B stores into the peer's slot -> s_waitcnt vmcnt(0)
//          -> per-peer flag store with __ATOMIC_RELEASE at
//          __HIP_MEMORY_SCOPE_SYSTEM (orders the data writes before the
//          flag for any acquiring observer on the destination device);
//  reader: relaxed system-scope spin on its LOCAL flags -> one
//          system-scope acquire fence -> plain reads of its local slots.
//  Slots and flags are double-buffered by epoch parity: parity p is
//  reused only at epoch e+2, and a rank can only reach the write phase
//  of e+2 after every rank completed e (its e+1 flags imply its e
//  kernel — including the reduce — retired on its in-order stream).
//
// Buffer layout (per rank, one hipMalloc block, IPC-shared):
//   [0,   64)  u64 peer_base[8]   (local copy of all ranks' bases)
//   [64,  72)  u64 epoch          (local; bumped by last-arriving block)
//   [72,  76)  u32 ticket         (local; re-armed each collective)
//   [76,  80)  u32 err            (sticky; host checks after sync)
//   [128, 2176) u64 flags[2][8][16]  (parity, writer, stripe)
//   [4096, ...) data slots [2][8][slot_bytes]
//
// A bounded spin (s_sleep backoff) turns a dead peer into a sticky err
// flag + clean return instead of a GPU hang (gpurun strike protection);
// the host raises on err at the next sync point.

#include <hip/hip_runtime.h>
#include <cstdint>

#include "common.h"

#define XC_MAX_WORLD 8
#define XC_MAX_STRIPES 16
#define XC_OFF_EPOCH 64
#define XC_OFF_TICKET 72
#define XC_OFF_ERR 76
#define XC_OFF_FLAGS 128
#define XC_OFF_DATA 4096

typedef unsigned long long u64t;

// MODE: 0 = all-reduce bf16 (fp32 accum), 1 = all-reduce fp32,
//       2 = all-gather (raw bytes; dst holds world*nbytes)
template <int MODE>
__global__ void __launch_bounds__(512)
k_xgmi_coll(char* __restrict__ dst, const char* __restrict__ src,
            char* __restrict__ mybase, int rank, int world, long nbytes,
            long slot_bytes, long spin_limit) {
  u64t* peer_base = (u64t*)mybase;
  u64t* epoch_ctr = (u64t*)(mybase + XC_OFF_EPOCH);
  uint32_t* ticket = (uint32_t*)(mybase + XC_OFF_TICKET);
  uint32_t* err = (uint32_t*)(mybase + XC_OFF_ERR);

  const u64t epoch = *epoch_ctr + 1;  // epochs start at 1; flags zeroed
  const int parity = (int)(epoch & 1);
  const int stripe = blockIdx.x;
  const int NS = gridDim.x;
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;

  // stripe byte range (16-B aligned; host guarantees nbytes % 16 == 0)
  const long per = ((nbytes / 16 + NS - 1) / NS) * 16;
  const long b0 = (long)stripe * per;
  const long b1 = (b0 + per < nbytes) ? b0 + per : nbytes;

  // ---- push my [b0,b1) to slot[parity][rank] of every rank (self incl.)
  const long slot_off =
      XC_OFF_DATA + ((long)parity * XC_MAX_WORLD + rank) * slot_bytes;
  for (int p = wave; p < world; p += nwaves) {
    char* slot = (char*)peer_base[p] + slot_off;
    for (long i = b0 + (long)lane * 16; i < b1; i += 64 * 16)
      *(u4v_*)(slot + i) = *(const u4v_*)(src + i);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    const long foff = ((long)parity * XC_MAX_WORLD + rank) * XC_MAX_STRIPES +
                      stripe;
    for (int p = 0; p < world; p++) {
      u64t* fl = (u64t*)((char*)peer_base[p] + XC_OFF_FLAGS) + foff;
      __hip_atomic_store(fl, epoch, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }

  // ---- wait for all writers' flags for MY stripe on MY buffer
  if (wave == 0 && lane < world) {
    u64t* fl = (u64t*)(mybase + XC_OFF_FLAGS) +
               ((long)parity * XC_MAX_WORLD + lane) * XC_MAX_STRIPES + stripe;
    long spins = 0;
    while (__hip_atomic_load(fl, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) < epoch) {
      if (++spins > spin_limit) {
        __hip_atomic_store(err, 1u, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        break;
      }
      __builtin_amdgcn_s_sleep(8);
    }
  }
  __syncthreads();
  if (__hip_atomic_load(err, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)) {
    // dead peer: leave dst untouched, still bump the epoch so state
    // stays consistent if the host decides to tear down gracefully
    __syncthreads();
    if (threadIdx.x == 0) {
      uint32_t t = __hip_atomic_fetch_add(ticket, 1, __ATOMIC_ACQ_REL,
                                          __HIP_MEMORY_SCOPE_AGENT);
      if (t == (uint32_t)NS - 1) { *ticket = 0; *epoch_ctr = epoch; }
    }
    return;
  }
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");  // slot data now visible

  // ---- reduce / gather from MY slots
  char* slots = mybase + XC_OFF_DATA + (long)parity * XC_MAX_WORLD * slot_bytes;
  if (MODE == 0) {
    for (long i = b0 + (long)threadIdx.x * 16; i < b1;
         i += (long)blockDim.x * 16) {
      float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
      for (int s = 0; s < world; s++) {
        s8v v = *(const s8v*)(slots + (long)s * slot_bytes + i);
#pragma unroll
        for (int j = 0; j < 8; j++) acc[j] += b2f(((u16*)&v)[j]);
      }
      u16 o[8];
#pragma unroll
      for (int j = 0; j < 8; j++) o[j] = f2b(acc[j]);
      *(s8v*)(dst + i) = *(s8v*)o;
    }
  } else if (MODE == 1) {
    for (long i = b0 + (long)threadIdx.x * 16; i < b1;
         i += (long)blockDim.x * 16) {
      f4v acc = {0.f, 0.f, 0.f, 0.f};
      for (int s = 0; s < world; s++) {
        f4v v = *(const f4v*)(slots + (long)s * slot_bytes + i);
        acc += v;
      }
      *(f4v*)(dst + i) = acc;
    }
  } else {
    for (int s = wave; s < world; s += nwaves) {
      for (long i = b0 + (long)lane * 16; i < b1; i += 64 * 16)
        *(u4v_*)(dst + (long)s * nbytes + i) =
            *(const u4v_*)(slots + (long)s * slot_bytes + i);
    }
  }

  // ---- epoch bump by the last-arriving block (graph-replay re-arm)
  __syncthreads();
  if (threadIdx.x == 0) {
    uint32_t t = __hip_atomic_fetch_add(ticket, 1, __ATOMIC_ACQ_REL,
                                        __HIP_MEMORY_SCOPE_AGENT);
    if (t == (uint32_t)NS - 1) { *ticket = 0; *epoch_ctr = epoch; }
  }
}

extern "C" hipError_t launch_xgmi_coll(void* dst, const void* src,
                                       void* mybase, int rank, int world,
                                       long nbytes, long slot_bytes, int mode,
                                       int nstripes, long spin_limit,
                                       hipStream_t stream) {
  if (nbytes % 16 != 0 || world > XC_MAX_WORLD ||
      nstripes > XC_MAX_STRIPES || nbytes > slot_bytes)
    return hipErrorInvalidValue;
  dim3 grid(nstripes);
#define XC_CASE(M)                                                         \
  hipLaunchKernelGGL((k_xgmi_coll<M>), grid, dim3(512), 0, stream,         \
                     (char*)dst, (const char*)src, (char*)mybase, rank,    \
                     world, nbytes, slot_bytes, spin_limit)
  if (mode == 0) XC_CASE(0);
  else if (mode == 1) XC_CASE(1);
  else if (mode == 2) XC_CASE(2);
  else return hipErrorInvalidValue;
#undef XC_CASE
  return hipGetLastError();
}

// ====================================================================
// host-side plumbing (raw hipMalloc so the base pointer is IPC-stable;
// torch's caching allocator hands out offsets into larger blocks)
// ====================================================================

extern "C" hipError_t xc_alloc(long bytes, void** out) {
  return hipMalloc(out, (size_t)bytes);
}

extern "C" hipError_t xc_free(void* p) { return hipFree(p); }

extern "C" hipError_t xc_memset(void* p, int v, long bytes) {
  return hipMemset(p, v, (size_t)bytes);
}

extern "C" hipError_t xc_h2d(void* dst, const void* src, long bytes) {
  return hipMemcpy(dst, src, (size_t)bytes, hipMemcpyHostToDevice);
}

extern "C" hipError_t xc_d2h(void* dst, const void* src, long bytes) {
  return hipMemcpy(dst, src, (size_t)bytes, hipMemcpyDeviceToHost);
}

extern "C" hipError_t xc_ipc_handle(void* ptr, void* out64) {
  return hipIpcGetMemHandle((hipIpcMemHandle_t*)out64, ptr);
}

extern "C" hipError_t xc_ipc_open(const void* handle64, void** out) {
  hipIpcMemHandle_t h;
  __builtin_memcpy(&h, handle64, sizeof(h));
  return hipIpcOpenMemHandle(out, h, hipIpcMemLazyEnablePeerAccess);
}

extern "C" hipError_t xc_ipc_close(void* ptr) {
  return hipIpcCloseMemHandle(ptr);
}
