// One-shot all-reduce over xGMI peer-to-peer for decode-size tensors.
//
// MI355X topology: the 8-GPU node is FULLY CONNECTED —
// 7 xGMI links per GPU at ~153 GB/s each.  A ring all-reduce moves
// 2·(N-1)/N·B bytes over ONE link serially plus 2(N-1) latency hops; a
// one-shot all-reduce has every rank read all peers' contributions
// DIRECTLY, moving (N-1)/N·B bytes per link in PARALLEL with one
// synchronization round.  For the ≤-few-MB tensors of decode-time
// tensor parallelism this is both lower latency and higher bandwidth
// than the ring, and it is hipGraph-capturable by construction (plain
// kernel launches, device-side epoch counters — no RCCL in the graph).
//
// Protocol per call (stream-ordered, one kernel):
//   phase 0: block b copies its slice of this rank's input into this
//            rank's shared buffer (double-buffered by call parity);
//   fence:   system-scope release, then block b writes its epoch into
//            slot [b][me] of EVERY rank's signal page;
//   spin:    block b polls its OWN signal page until all ranks' [b][r]
//            have reached the epoch (relaxed system loads + one system
//            acquire), with a bounded spin that sets an error word
//            instead of hanging the device;
//   phase 1: block b sums slice b from all ranks' shared buffers
//            (bf16 -> fp32 accumulate) and writes the result.
//
// Epochs advance in device memory (one counter per block), so graph
// replays are self-synchronizing.  Double buffering by epoch parity
// makes a one-call skew between ranks safe: a rank can only be writing
// parity p when every rank has finished reading parity p from the call
// before last (the flag handshake orders it).
//
// Cross-GPU visibility notes (MI355X_MICROARCH.md §Workgroup dispatch):
// peer writes land over xGMI; we use SYSTEM-scope atomics for flags and
// an explicit s_waitcnt before the release (the ROCm 7.2 compiler can
// drop the post-release wait when the scoreboard looks clean — guide
// §6 G16 pitfall 12; the asm form is immune).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace igw {

constexpr int AR_MAX_RANKS = 8;
constexpr int AR_BLOCKS = 64;
constexpr int AR_THREADS = 256;
// spin bound: ~2.4 GHz, a few hundred ms worth of polls — long enough
// for any real skew, short enough to never look like a hung GPU
constexpr long long AR_SPIN_LIMIT = 20LL * 1000 * 1000;

struct OneShotSig {
  // [block][rank]: written by rank `rank` (its block `block`), on every
  // rank's page; polled by the owner of the page
  unsigned int flags[AR_BLOCKS][AR_MAX_RANKS];
  unsigned int epochs[AR_BLOCKS];   // private per-owner call counters
  unsigned int error;               // set on spin timeout
};

struct PtrTable {
  const uint16_t* data[AR_MAX_RANKS];
  OneShotSig* sig[AR_MAX_RANKS];
};

typedef int igw_v4i __attribute__((ext_vector_type(4)));
// aux bits for buffer ops: sc1 = write-through / no-allocate — shared
// data goes straight to the coherence point and reads bypass stale
// per-XCD L2 lines (guide §6 G16 R1: the valid all-sc1 protocol form;
// plain stores + fences measured intermittently stale cross-XCD here)
constexpr int AUX_SC1 = 16;

__global__ void __launch_bounds__(AR_THREADS)
oneshot_allreduce_kernel(uint16_t* __restrict__ out,
                         const uint16_t* __restrict__ inp,
                         PtrTable tab, int rank, int world, long numel,
                         long buf_stride /* elements per parity half */) {
  const int b = blockIdx.x;
  __shared__ unsigned int s_epoch;
  OneShotSig* my_sig = tab.sig[rank];
  if (threadIdx.x == 0) {
    if (b == 0) my_sig->error = 0;   // a past timeout is not this call's
    s_epoch = my_sig->epochs[b] + 1;
    my_sig->epochs[b] = s_epoch;
  }
  __syncthreads();
  const unsigned int epoch = s_epoch;
  const long par_off = (epoch & 1) ? buf_stride : 0;

  // slice owned by this block (grid-stride inside the slice keeps the
  // phases' element->block map identical)
  const long per_block = (numel + gridDim.x - 1) / gridDim.x;
  const long lo = (long)b * per_block;
  const long hi0 = lo + per_block;
  const long hi = hi0 < numel ? hi0 : numel;
  // vectors cover [lo, tail); the ragged remainder [tail, hi) is scalar
  const long tail = hi > lo ? lo + ((hi - lo) / 8) * 8 : lo;
  uint16_t* my_buf = const_cast<uint16_t*>(tab.data[rank]) + par_off;
  // write-through descriptor over my parity half (32-bit offsets)
  const auto my_rsrc = __builtin_amdgcn_make_buffer_rsrc(
      (void*)my_buf, 0, (unsigned)(numel * 2), 0x00020000);

  // ---- phase 0: my slice -> my shared buffer, sc1 write-through ------
  {
    for (long i = lo + (long)threadIdx.x * 8; i + 8 <= hi;
         i += (long)AR_THREADS * 8) {
      const uint4 v = *reinterpret_cast<const uint4*>(inp + i);
      __builtin_amdgcn_raw_buffer_store_b128(
          *reinterpret_cast<const igw_v4i*>(&v), my_rsrc,
          (unsigned)(i * 2), 0, AUX_SC1);
    }
    if (threadIdx.x == 0) {
      for (long j = tail; j < hi; ++j) {
        __builtin_amdgcn_raw_buffer_store_b16(
            (short)inp[j], my_rsrc, (unsigned)(j * 2), 0, AUX_SC1);
      }
    }
  }
  // EVERY storing wave drains its write-through stores to the coherence
  // point before any lane publishes the flag (guide §6 G16 R1)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // ---- signal: stamp epoch on every rank's page ----------------------
  // RELEASE stores (system scope): the conservative documented form —
  // the payload is already write-through, but release-ordering the flag
  // also covers any store still in flight in this lane's queue
  if (threadIdx.x < (unsigned)world) {
    __hip_atomic_store(&tab.sig[threadIdx.x]->flags[b][rank], epoch,
                       __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
  }

  // ---- spin: wait for every rank's slice b (bounded) -----------------
  if (threadIdx.x == 0) {
    long long spins = 0;
    for (int r = 0; r < world; ++r) {
      for (;;) {
        // poll MY OWN page with agent scope (the guide's measured
        // wait_eq form: sc1 load, no L2 allocate); the remote writer's
        // store is system-scope and lands here over the fabric
        // ACQUIRE poll, system scope: "polling with ACQUIRE loads ->
        // correct, 2-3x slower per hop" (guide §Workgroup dispatch) —
        // the flag page is 64 words, the hop cost is irrelevant next
        // to the payload; relaxed polls measured rare indefinite
        // staleness here (per-XCD L2)
        unsigned int f = __hip_atomic_load(
            &my_sig->flags[b][r], __ATOMIC_ACQUIRE,
            __HIP_MEMORY_SCOPE_SYSTEM);
        if ((int)(f - epoch) >= 0) break;
        if (++spins > AR_SPIN_LIMIT) {
          // encode what we were waiting for: block, rank, last value
          __hip_atomic_store(&my_sig->error,
                             0x80000000u | ((unsigned)b << 20)
                                 | ((unsigned)r << 16) | (f & 0xffffu),
                             __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
          return;  // poisoned: host checks the error word
        }
        __builtin_amdgcn_s_sleep(2);
      }
    }
    // one unconditional system acquire before the sum: every rank's
    // DATA slice must be re-fetched past any stale L2 lines left from
    // two calls ago (the same parity half)
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
  }
  __syncthreads();

  // ---- phase 1: sum slice b from all ranks (sc1 loads) -> out --------
  {
    __amdgpu_buffer_rsrc_t rsrcs[AR_MAX_RANKS];
    for (int r = 0; r < world; ++r) {
      rsrcs[r] = __builtin_amdgcn_make_buffer_rsrc(
          (void*)(tab.data[r] + par_off), 0, (unsigned)(numel * 2),
          0x00020000);
    }
    for (long i = lo + (long)threadIdx.x * 8; i + 8 <= hi;
         i += (long)AR_THREADS * 8) {
      float acc[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) acc[k] = 0.f;
      for (int r = 0; r < world; ++r) {
        const igw_v4i v = __builtin_amdgcn_raw_buffer_load_b128(
            rsrcs[r], (unsigned)(i * 2), 0, AUX_SC1);
        const uint16_t* h = reinterpret_cast<const uint16_t*>(&v);
#pragma unroll
        for (int k = 0; k < 8; ++k) acc[k] += bf16_to_f32(h[k]);
      }
      uint4 o;
      uint16_t* oh = reinterpret_cast<uint16_t*>(&o);
#pragma unroll
      for (int k = 0; k < 8; ++k) oh[k] = f32_to_bf16(acc[k]);
      *reinterpret_cast<uint4*>(out + i) = o;
    }
    if (threadIdx.x == 0) {
      for (long j = tail; j < hi; ++j) {
        float acc = 0.f;
        for (int r = 0; r < world; ++r) {
          const short hv = __builtin_amdgcn_raw_buffer_load_b16(
              rsrcs[r], (unsigned)(j * 2), 0, AUX_SC1);
          acc += bf16_to_f32((uint16_t)hv);
        }
        out[j] = f32_to_bf16(acc);
      }
    }
  }
}

// --------------------------------------------------------------------------
// host API
// --------------------------------------------------------------------------
int64_t ar_alloc(int64_t bytes, bool finegrained) {
  // signal pages are FINE-GRAINED (coherent during kernel execution —
  // cross-device flag polling needs it; coarse-grained device memory is
  // only coherent at dispatch boundaries).  Data buffers stay
  // coarse-grained: the release/acquire fences order them, and coarse
  // remote reads stream over xGMI at full rate.
  void* p = nullptr;
  hipError_t err;
  if (finegrained) {
    err = hipExtMallocWithFlags(&p, bytes, hipDeviceMallocFinegrained);
  } else {
    err = hipMalloc(&p, bytes);
  }
  TORCH_CHECK(err == hipSuccess, "ar_alloc failed: ",
              hipGetErrorString(err));
  TORCH_CHECK(hipMemset(p, 0, bytes) == hipSuccess);
  return reinterpret_cast<int64_t>(p);
}

void ar_free(int64_t ptr) {
  hipFree(reinterpret_cast<void*>(ptr));
}

py::bytes ar_get_handle(int64_t ptr) {
  hipIpcMemHandle_t h;
  auto err = hipIpcGetMemHandle(&h, reinterpret_cast<void*>(ptr));
  TORCH_CHECK(err == hipSuccess, "hipIpcGetMemHandle failed: ",
              hipGetErrorString(err));
  return py::bytes(reinterpret_cast<const char*>(&h), sizeof(h));
}

int64_t ar_open_handle(py::bytes handle) {
  std::string s = handle;
  TORCH_CHECK(s.size() == sizeof(hipIpcMemHandle_t),
              "bad IPC handle size");
  hipIpcMemHandle_t h;
  memcpy(&h, s.data(), sizeof(h));
  void* p = nullptr;
  auto err = hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(err == hipSuccess, "hipIpcOpenMemHandle failed: ",
              hipGetErrorString(err));
  return reinterpret_cast<int64_t>(p);
}

void ar_close_handle(int64_t ptr) {
  hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr));
}

int64_t ar_sig_bytes() { return (int64_t)sizeof(OneShotSig); }

int64_t ar_check_error(int64_t sig_ptr) {
  unsigned int e = 0;
  hipMemcpy(&e, &reinterpret_cast<OneShotSig*>(sig_ptr)->error,
            sizeof(e), hipMemcpyDeviceToHost);
  return (int64_t)e;
}

py::bytes ar_dump(int64_t ptr, int64_t nbytes) {
  std::string s(nbytes, '\0');
  hipMemcpy(&s[0], reinterpret_cast<void*>(ptr), nbytes,
            hipMemcpyDeviceToHost);
  return py::bytes(s);
}

void oneshot_allreduce(torch::Tensor inout, int64_t rank, int64_t world,
                       std::vector<int64_t> data_ptrs,
                       std::vector<int64_t> sig_ptrs,
                       int64_t buf_stride_elems) {
  TORCH_CHECK(inout.is_cuda() && inout.is_contiguous());
  TORCH_CHECK(inout.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(world >= 1 && world <= AR_MAX_RANKS);
  TORCH_CHECK((int64_t)data_ptrs.size() >= world);
  const long numel = (long)inout.numel();
  TORCH_CHECK(numel <= buf_stride_elems,
              "tensor larger than the one-shot buffer");
  PtrTable tab{};
  for (int r = 0; r < world; ++r) {
    tab.data[r] = reinterpret_cast<const uint16_t*>(data_ptrs[r]);
    tab.sig[r] = reinterpret_cast<OneShotSig*>(sig_ptrs[r]);
  }
  uint16_t* ptr = reinterpret_cast<uint16_t*>(inout.data_ptr());
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(oneshot_allreduce_kernel, dim3(AR_BLOCKS),
                     dim3(AR_THREADS), 0, stream, ptr, ptr, tab,
                     (int)rank, (int)world, numel,
                     (long)buf_stride_elems);
}

}  // namespace igw
