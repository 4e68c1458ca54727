// One-shot direct all-reduce over xGMI for decode-sized messages.
//
// Ring all-reduce moves 2*(N-1)/N of the payload over ONE link per hop
// with N latency hops; on an MI355X node the 8 GPUs are fully connected
// by 7 point-to-point xGMI links per GPU, so for the
// small activation tensors of TP decode (hidden*batch bf16, <= ~1 MB)
// the right shape is a single-shot exchange: every rank stages its
// contribution locally, publishes a sequence flag into every peer's
// flag array (push), then pulls all N staged buffers over all links at
// once and reduces. Two tiny kernels per call, both hipGraph-capturable
// (device-resident sequence counter, no host sync).
//
// Cross-device ordering: payload stores are made visible with a
// system-scope release fence before the remote flag store; consumers
// poll their LOCAL flag array (peers push) with acquire loads. IPC
// mapping of peer buffers uses hipIpc (dmabuf mode in this image).
#include "common.h"
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <vector>

#define CUR_STREAM c10::hip::getCurrentHIPStream().stream()
#define AR_MAX_WORLD 8

// shared-buffer layout (per rank, one hipMalloc):
//   [0]                 uint64 flags[AR_MAX_WORLD]   (peers push here)
//   [256]               uint64 seq                   (local only)
//   [512 ...]           payload staging A (max_bytes)
//   [512+max ...]       payload staging B (max_bytes)
// Staging is parity double-buffered on seq: a rank may legally be ONE
// call ahead of a slow peer (it publishes call k's flag before its own
// reduce of k completes), so call k+1's staging must not overwrite call
// k's payload. Two calls ahead is impossible: starting call k+2 requires
// having observed every peer's k+1 flag, which a peer only pushes after
// its reduce of k fully retired (stream order on that peer).
#define AR_FLAGS_OFF 0
#define AR_SEQ_OFF 256
#define AR_DATA_OFF 512

struct ARState {
  int world = 0;
  int rank = -1;
  size_t max_bytes = 0;
  void* bufs[AR_MAX_WORLD] = {nullptr};  // bufs[rank] is the local alloc
  bool peer_mapped[AR_MAX_WORLD] = {false};
};
static ARState g_ar;

__host__ __device__ static inline unsigned long long* ar_flags(void* base) {
  return (unsigned long long*)((char*)base + AR_FLAGS_OFF);
}
__host__ __device__ static inline unsigned long long* ar_seq(void* base) {
  return (unsigned long long*)((char*)base + AR_SEQ_OFF);
}
__host__ __device__ static inline __hip_bfloat16* ar_data(void* base, unsigned long long seq,
                                      size_t max_bytes) {
  return (__hip_bfloat16*)((char*)base + AR_DATA_OFF
                           + (seq & 1) * max_bytes);
}

// K0: bump the call sequence number (1 thread; keeps K1/K2 reads stable)
__global__ void ar_bump_kernel(unsigned long long* seq) {
  __hip_atomic_store(seq, __hip_atomic_load(seq, __ATOMIC_RELAXED,
                                            __HIP_MEMORY_SCOPE_AGENT) + 1,
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// K1: copy the input into the seq-parity staging buffer. Stores are
// 8-byte system-scope (write-through sc1): per-XCD L2s are not snooped
// by remote xGMI reads, so staged payload must not sit dirty in L2 when
// the peer pulls it — a single fence in K2 block 0 would only write back
// ONE XCD's L2 (MI355X_MICROARCH.md §Workgroup dispatch).
__global__ void ar_stage_kernel(const __hip_bfloat16* __restrict__ in,
                                void* local_buf, unsigned long long* seq,
                                long n, long max_bytes) {
  const unsigned long long s = __hip_atomic_load(
      seq, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  __hip_bfloat16* stage = ar_data(local_buf, s, (size_t)max_bytes);
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long nv = n / 4;  // 8-byte (4 x bf16) write-through granules
  const unsigned long long* src = (const unsigned long long*)in;
  unsigned long long* dst = (unsigned long long*)stage;
  for (long i = i0; i < nv; i += stride)
    __hip_atomic_store(&dst[i], src[i], __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_SYSTEM);
  for (long i = nv * 4 + i0; i < n; i += stride) {
    unsigned short v = *(const unsigned short*)&in[i];
    __hip_atomic_store((unsigned short*)&stage[i], v, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

// K2: publish my seq to every peer's flag slot, wait for all peers'
// flags to reach seq, then pull + reduce all staged buffers.
template <int WORLD>
__global__ void ar_reduce_kernel(void* const* bufs, int my_rank,
                                 __hip_bfloat16* __restrict__ out, long n,
                                 long max_bytes) {
  __shared__ unsigned long long s_seq;
  unsigned long long* my_seq = ar_seq(bufs[my_rank]);
  if (threadIdx.x == 0) {
    s_seq = __hip_atomic_load(my_seq, __ATOMIC_RELAXED,
                              __HIP_MEMORY_SCOPE_AGENT);
  }
  __syncthreads();
  const unsigned long long seq = s_seq;
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    // K1's write-through payload stores all retired at its kernel
    // boundary; restate the drain where the compiler cannot drop it,
    // then push the flag to every rank
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __threadfence_system();
#pragma unroll
    for (int r = 0; r < WORLD; ++r) {
      __hip_atomic_store(&ar_flags(bufs[r])[my_rank], seq, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_SYSTEM);
    }
  }
  // every block waits until ALL ranks' flags (pushed into OUR local flag
  // array) reach seq
  if (threadIdx.x < WORLD) {
    volatile unsigned long long* f = &ar_flags(bufs[my_rank])[threadIdx.x];
    while (__hip_atomic_load((unsigned long long*)f, __ATOMIC_ACQUIRE,
                             __HIP_MEMORY_SCOPE_SYSTEM) < seq) {
      __builtin_amdgcn_s_sleep(1);
    }
  }
  __syncthreads();
  // one-shot pull: each element sums WORLD staged copies (7 remote pulls
  // ride 7 xGMI links concurrently)
  const __hip_bfloat16* stage[WORLD];
#pragma unroll
  for (int r = 0; r < WORLD; ++r)
    stage[r] = ar_data(bufs[r], seq, (size_t)max_bytes);
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  const long nv = n / 4;
  for (long i = i0; i < nv; i += stride) {
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int r = 0; r < WORLD; ++r) {
      // sc1 load: bypasses L1, pairs with the producers' sc1 stores
      const unsigned long long v = __hip_atomic_load(
          &((const unsigned long long*)stage[r])[i], __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_SYSTEM);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[j] += bf16_to_f32((unsigned short)(v >> (16 * j)));
    }
    unsigned long long o = 0;
#pragma unroll
    for (int j = 0; j < 4; ++j)
      o |= (unsigned long long)f32_to_bf16(acc[j]) << (16 * j);
    ((unsigned long long*)out)[i] = o;
  }
  for (long i = nv * 4 + i0; i < n; i += stride) {
    float acc = 0.f;
#pragma unroll
    for (int r = 0; r < WORLD; ++r) {
      unsigned short v = __hip_atomic_load(
          (const unsigned short*)&stage[r][i], __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_SYSTEM);
      acc += bf16_to_f32(v);
    }
    *(unsigned short*)&out[i] = f32_to_bf16(acc);
  }
}

// device-side array of peer base pointers (uploaded once at connect)
static void** g_dev_bufs = nullptr;

std::vector<uint8_t> oneshot_init(long world, long rank, long max_bytes) {
  TORCH_CHECK(world >= 2 && world <= AR_MAX_WORLD, "world must be 2..8");
  TORCH_CHECK(rank >= 0 && rank < world);
  g_ar.world = (int)world;
  g_ar.rank = (int)rank;
  g_ar.max_bytes = (size_t)max_bytes;
  void* buf = nullptr;
  size_t total = AR_DATA_OFF + 2 * (size_t)max_bytes;  // parity buffers
  // fine-grained device memory: remote xGMI accesses are coherent at
  // system scope (RCCL's own transport buffers use the same attribute);
  // coarse-grained IPC memory gives no cross-device ordering guarantees
  hipError_t rc = hipExtMallocWithFlags(&buf, total,
                                        hipDeviceMallocFinegrained);
  TORCH_CHECK(rc == hipSuccess, "oneshot finegrained alloc failed: ",
              hipGetErrorString(rc));
  TORCH_CHECK(hipMemset(buf, 0, AR_DATA_OFF) == hipSuccess);
  g_ar.bufs[rank] = buf;
  hipIpcMemHandle_t h;
  TORCH_CHECK(hipIpcGetMemHandle(&h, buf) == hipSuccess,
              "hipIpcGetMemHandle failed");
  return std::vector<uint8_t>((uint8_t*)&h, (uint8_t*)&h + sizeof(h));
}

void oneshot_connect(std::vector<std::vector<uint8_t>> handles) {
  TORCH_CHECK((int)handles.size() == g_ar.world, "need one handle per rank");
  for (int r = 0; r < g_ar.world; ++r) {
    if (r == g_ar.rank) continue;
    TORCH_CHECK(handles[r].size() == sizeof(hipIpcMemHandle_t));
    hipIpcMemHandle_t h;
    memcpy(&h, handles[r].data(), sizeof(h));
    void* p = nullptr;
    TORCH_CHECK(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess)
                == hipSuccess, "hipIpcOpenMemHandle failed for rank ", r);
    g_ar.bufs[r] = p;
    g_ar.peer_mapped[r] = true;
  }
  if (g_dev_bufs == nullptr)
    TORCH_CHECK(hipMalloc(&g_dev_bufs, AR_MAX_WORLD * sizeof(void*))
                == hipSuccess);
  TORCH_CHECK(hipMemcpy(g_dev_bufs, g_ar.bufs,
                        AR_MAX_WORLD * sizeof(void*),
                        hipMemcpyHostToDevice) == hipSuccess);
}

bool oneshot_ready() {
  if (g_ar.world == 0 || g_dev_bufs == nullptr) return false;
  for (int r = 0; r < g_ar.world; ++r)
    if (g_ar.bufs[r] == nullptr) return false;
  return true;
}

long oneshot_max_bytes() { return (long)g_ar.max_bytes; }

void oneshot_allreduce_(torch::Tensor t) {
  TORCH_CHECK(oneshot_ready(), "oneshot allreduce not initialized");
  TORCH_CHECK(t.is_cuda() && t.dtype() == torch::kBFloat16 && t.is_contiguous());
  const long n = t.numel();
  TORCH_CHECK((size_t)n * 2 <= g_ar.max_bytes, "message exceeds staging");
  auto stream = CUR_STREAM;
  void* local = g_ar.bufs[g_ar.rank];
  const int threads = 256;
  const int blocks = (int)std::min<long>((n / 8 + threads - 1) / threads + 1, 256);
  hipLaunchKernelGGL(ar_bump_kernel, dim3(1), dim3(1), 0, stream,
                     ar_seq(local));
  hipLaunchKernelGGL(ar_stage_kernel, dim3(blocks), dim3(threads), 0, stream,
                     (const __hip_bfloat16*)t.data_ptr(), local,
                     ar_seq(local), n, (long)g_ar.max_bytes);
#define AR_CASE(W)                                                         \
  case W:                                                                  \
    hipLaunchKernelGGL((ar_reduce_kernel<W>), dim3(blocks), dim3(threads), \
                       0, stream, (void* const*)g_dev_bufs, g_ar.rank,     \
                       (__hip_bfloat16*)t.data_ptr(), n,                   \
                       (long)g_ar.max_bytes);                              \
    break;
  switch (g_ar.world) {
    AR_CASE(2) AR_CASE(3) AR_CASE(4) AR_CASE(5)
    AR_CASE(6) AR_CASE(7) AR_CASE(8)
    default: TORCH_CHECK(false, "bad world");
  }
#undef AR_CASE
  HIP_CHECK_KERNEL();
}

void oneshot_destroy() {
  for (int r = 0; r < AR_MAX_WORLD; ++r) {
    if (g_ar.peer_mapped[r] && g_ar.bufs[r]) {
      hipIpcCloseMemHandle(g_ar.bufs[r]);
    } else if (r == g_ar.rank && g_ar.bufs[r]) {
      hipFree(g_ar.bufs[r]);
    }
    g_ar.bufs[r] = nullptr;
    g_ar.peer_mapped[r] = false;
  }
  g_ar.world = 0;
  g_ar.rank = -1;
}
