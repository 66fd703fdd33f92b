// _hipstore — native core of torchstore_amd for MI355X (gfx950, CDNA4).
//
// Provides (python surface wrapped by torchstore_amd/ops/gpu.py):
//   * HIP IPC:  ipc_export / ipc_open / ipc_close — hipIpcMemHandle_t at
//     caching-allocator block granularity (base via hipMemGetAddressRange,
//     descriptor carries the byte offset), replacing the reference's
//     ibverbs RDMA registration (torchstore transport/monarch_rdma.py).
//   * copy_batch — batched one-sided bulk copies (hipMemcpyPeerAsync /
//     DtoD) striped round-robin over a per-device pool of dedicated HIP
//     streams so concurrent transfers to different peers aggregate xGMI
//     links (7 x ~153 GB/s per GPU); synchronizes the streams it used.
//   * copy_slices — K1/K2: one kernel launch copying N strided slices
//     (gather: strided->contiguous, scatter: contiguous->strided, or
//     strided->strided), replacing per-slice torch copies in reshard
//     pack/unpack (reference hot loops: storage_volume.py:220-277,
//     utils.py:199-212, direct_weight_sync.py:350).
//   * cast_copy — K3: fused dtype cast + pack (f32<->bf16/f16), one HBM
//     read + one write, vectorized 16B/lane (reference:
//     state_dict_utils.py:177-189 casts every floating param per sync).
//   * host_register / host_unregister — pin SHM pages for DMA.
//
// Deliberately torch-ABI-free: tensors cross as raw device pointers +
// geometry, so the module builds with plain hipcc + pybind11 and never
// drifts against libtorch.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <mutex>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      (void)hipGetLastError(); /* clear the sticky error so the caller's */    \
      /* process (torch's own error checks) is not poisoned */                 \
      throw std::runtime_error(std::string(#expr) + " failed: " +              \
                               hipGetErrorString(_e));                         \
    }                                                                          \
  } while (0)

// ---------------------------------------------------------------------------
// stream pools: dedicated streams per device for bulk copies
// ---------------------------------------------------------------------------

static constexpr int kStreamsPerDevice = 8;

struct DevicePool {
  std::vector<hipStream_t> streams;
  // pinned host staging for slice descriptors + device copy of them
  void* h_desc = nullptr;
  void* d_desc = nullptr;
  size_t desc_cap = 0;
  // guards h_desc reuse: recorded after each staging H2D enqueue
  hipEvent_t desc_evt = nullptr;
};

static std::mutex g_mutex;
static std::unordered_map<int, DevicePool> g_pools;

static DevicePool& pool_for(int device) {
  std::lock_guard<std::mutex> lock(g_mutex);
  auto it = g_pools.find(device);
  if (it != g_pools.end()) return it->second;
  HIP_CHECK(hipSetDevice(device));
  DevicePool p;
  p.streams.resize(kStreamsPerDevice);
  for (int i = 0; i < kStreamsPerDevice; ++i) {
    HIP_CHECK(hipStreamCreateWithFlags(&p.streams[i], hipStreamNonBlocking));
  }
  return g_pools.emplace(device, std::move(p)).first->second;
}

static void ensure_desc_capacity(DevicePool& p, int device, size_t bytes) {
  if (p.desc_cap >= bytes) return;
  size_t cap = bytes * 2 + 4096;
  HIP_CHECK(hipSetDevice(device));
  if (p.h_desc) HIP_CHECK(hipHostFree(p.h_desc));
  if (p.d_desc) HIP_CHECK(hipFree(p.d_desc));
  HIP_CHECK(hipHostMalloc(&p.h_desc, cap, hipHostMallocDefault));
  HIP_CHECK(hipMalloc(&p.d_desc, cap));
  p.desc_cap = cap;
}

// ---------------------------------------------------------------------------
// HIP IPC
// ---------------------------------------------------------------------------

// Export handles are cached per allocator block (hipIpcGetMemHandle does a
// dmabuf export ioctl — tens of ms — while hipMemGetAddressRange is cheap).
// Same design as the reference's (data_ptr, nbytes)-keyed RdmaMemory cache
// (torchstore torchcomms/cache.py:150-187).  A cached handle goes stale
// only when its block is returned to the OS (torch.cuda.empty_cache /
// OOM-retry release) and the same base address is re-allocated; callers
// therefore pass an allocator GENERATION (the caching allocator's
// segment-freed counter): a changed generation flushes that device's
// cache before lookup — no manual clear call needed for correctness.
struct ExportCache {
  long long gen = -(1ll << 62);
  std::unordered_map<uintptr_t, std::string> map;
};
static std::mutex g_export_mutex;
static std::unordered_map<int, ExportCache> g_export_caches;

// Returns (handle_bytes, offset_in_block, block_size).  Callers MUST route
// blocks >= 2 GiB through the chunked-staging path: hipIpcOpenMemHandle of
// a >=2^31-byte dmabuf hangs on this platform (measured; the export itself
// succeeds, the peer's import never returns).
static py::tuple ipc_export(uintptr_t ptr, int device, long long generation) {
  HIP_CHECK(hipSetDevice(device));
  void* base = nullptr;
  size_t size = 0;
  HIP_CHECK(hipMemGetAddressRange(&base, &size, reinterpret_cast<void*>(ptr)));
  uintptr_t base_u = reinterpret_cast<uintptr_t>(base);
  std::string handle_str;
  if (generation >= 0) {  // negative generation = caller opts out of caching
    std::lock_guard<std::mutex> lock(g_export_mutex);
    ExportCache& c = g_export_caches[device];
    if (generation != c.gen) {
      c.map.clear();
      c.gen = generation;
    }
    auto it = c.map.find(base_u);
    if (it != c.map.end()) {
      handle_str = it->second;
    }
  }
  if (handle_str.empty()) {
    hipIpcMemHandle_t handle;
    HIP_CHECK(hipIpcGetMemHandle(&handle, base));
    handle_str.assign(reinterpret_cast<const char*>(&handle), sizeof(handle));
    if (generation >= 0) {
      std::lock_guard<std::mutex> lock(g_export_mutex);
      g_export_caches[device].map.emplace(base_u, handle_str);
    }
  }
  return py::make_tuple(py::bytes(handle_str),
                        static_cast<uint64_t>(ptr - base_u),
                        static_cast<uint64_t>(size));
}

static void ipc_export_cache_clear() {
  std::lock_guard<std::mutex> lock(g_export_mutex);
  g_export_caches.clear();
}

static uintptr_t ipc_open(py::bytes handle_bytes, int local_device,
                          int src_device) {
  std::string raw = handle_bytes;
  if (raw.size() != sizeof(hipIpcMemHandle_t)) {
    throw std::runtime_error("bad ipc handle size");
  }
  hipIpcMemHandle_t handle;
  std::memcpy(&handle, raw.data(), sizeof(handle));
  HIP_CHECK(hipSetDevice(local_device));
  void* ptr = nullptr;
  HIP_CHECK(hipIpcOpenMemHandle(&ptr, handle, hipIpcMemLazyEnablePeerAccess));
  (void)src_device;
  return reinterpret_cast<uintptr_t>(ptr);
}

static void ipc_close(uintptr_t base, int local_device) {
  HIP_CHECK(hipSetDevice(local_device));
  HIP_CHECK(hipIpcCloseMemHandle(reinterpret_cast<void*>(base)));
}

// ---------------------------------------------------------------------------
// batched bulk copies over xGMI / HBM
// ---------------------------------------------------------------------------

// copies: (dst_ptr, dst_dev, src_ptr, src_dev, nbytes)
//
// Same-device copies batch into ONE copy_slices kernel launch per device
// (291 sequential hipMemcpyAsync enqueues cost ~20 us each on the host —
// the kernel path replaces them with one dispatch).  Cross-device copies
// keep hipMemcpyPeerAsync striped round-robin over the stream pool so
// transfers to different peers ride different xGMI links concurrently.
// Definition follows the slice kernel below.
static void copy_batch(
    const std::vector<std::tuple<uintptr_t, int, uintptr_t, int, uint64_t>>&
        copies);

// copies: (dst_ptr, dst_dev, dpitch, src_ptr, src_dev, spitch, width, height)
// — strided (pitched) one-sided reads/writes: moves ONLY the overlap bytes
// of a reshard instead of whole remote shards (the reference always reads
// the full source shard, direct_weight_sync.py:280-314).
static void copy_batch_2d(
    const std::vector<std::tuple<uintptr_t, int, uint64_t, uintptr_t, int,
                                 uint64_t, uint64_t, uint64_t>>& copies) {
  if (copies.empty()) return;
  std::vector<hipStream_t> used;
  int i = 0;
  for (const auto& c : copies) {
    uintptr_t dst = std::get<0>(c);
    int dst_dev = std::get<1>(c);
    uint64_t dpitch = std::get<2>(c);
    uintptr_t src = std::get<3>(c);
    uint64_t spitch = std::get<5>(c);
    uint64_t width = std::get<6>(c);
    uint64_t height = std::get<7>(c);
    DevicePool& p = pool_for(dst_dev);
    hipStream_t s = p.streams[i++ % kStreamsPerDevice];
    HIP_CHECK(hipSetDevice(dst_dev));
    HIP_CHECK(hipMemcpy2DAsync(reinterpret_cast<void*>(dst), dpitch,
                               reinterpret_cast<void*>(src), spitch, width,
                               height, hipMemcpyDefault, s));
    used.push_back(s);
  }
  for (hipStream_t s : used) HIP_CHECK(hipStreamSynchronize(s));
}

// ---------------------------------------------------------------------------
// K1/K2 — batched strided slice copy
// ---------------------------------------------------------------------------
//
// Each slice: rows of `row_bytes` contiguous bytes; row r's source/dest
// offsets come from the outer-dims multi-index against byte strides.
// Work unit = one TILE_BYTES span of one row; blocks grid-stride over the
// global unit list and binary-search their slice in a prefix array.

static constexpr int kMaxDims = 7;       // outer dims (innermost is the row)
// tile size per launch: big batches amortize per-tile overhead with 128 KiB
// tiles; smaller ones keep 16 KiB tiles so enough waves stay busy
// (HIPSTORE_TILE overrides for experiments; read per call so tuning
// sweeps can vary it without a fresh process)
// defaults from the hardware tile x grid sweep (profiles/kernel_tune.log):
// 32 KiB tiles + an 8192-block grid cap won the flat bulk pattern (+5%
// over 128 KiB/2048); small batches keep 16 KiB tiles
static uint32_t pick_tile(uint64_t total_bytes) {
  const char* e = getenv("HIPSTORE_TILE");
  uint32_t forced = e ? (uint32_t)atoi(e) : 0u;
  if (forced >= 4096) return forced;
  return total_bytes > (512ull << 20) ? 32768u : 16384u;
}

// grid cap for the slice kernel (blocks); HIPSTORE_GRID overrides
static uint32_t pick_grid_cap() {
  const char* e = getenv("HIPSTORE_GRID");
  uint32_t forced = e ? (uint32_t)atoi(e) : 0u;
  return forced >= 64 ? forced : 8192u;
}

// row-packed units span this many bytes of small rows regardless of the
// launch tile (64 KiB won the scatter sweep at every grid size)
static constexpr uint32_t kRowPackSpan = 65536u;

// the cast kernel prefers a SMALLER grid (1024 beat 2048 by ~8% in the
// sweep — fewer blocks, longer per-block streams)
static uint32_t pick_cast_grid_cap() {
  const char* e = getenv("HIPSTORE_CAST_GRID");
  uint32_t forced = e ? (uint32_t)atoi(e) : 0u;
  return forced >= 64 ? forced : 1024u;
}

struct SliceDesc {
  uintptr_t src;
  uintptr_t dst;
  uint64_t rows;            // product of outer dims
  uint64_t units_prefix;    // exclusive prefix sum of units
  uint32_t row_bytes;
  uint32_t tiles_per_row;
  uint32_t ndim;            // number of outer dims
  // >1 = row-packed: one unit covers this many consecutive rows (small
  // rows, ndim<=1, fully 16B-aligned — the reshard scatter case, where a
  // per-row unit would pay desc-load + search overhead per ~1 KB of work)
  uint32_t rows_per_unit;
  uint64_t shape[kMaxDims];       // outer dims, innermost-last
  int64_t src_stride[kMaxDims];   // byte strides of outer dims
  int64_t dst_stride[kMaxDims];
};

__device__ __forceinline__ void row_offsets(const SliceDesc& d, uint64_t row,
                                            int64_t& soff, int64_t& doff) {
  soff = 0;
  doff = 0;
  uint64_t rem = row;
  for (int i = (int)d.ndim - 1; i >= 0; --i) {
    uint64_t idx = rem % d.shape[i];
    rem /= d.shape[i];
    soff += (int64_t)idx * d.src_stride[i];
    doff += (int64_t)idx * d.dst_stride[i];
  }
}

// One WAVE per work unit (a <=16 KiB span of one row): a 256-thread block
// runs 4 independent waves, so rows never serialize behind each other
// inside a block, and the x4-unrolled 16B path keeps 4 loads in flight
// per lane (64 lanes x 16B x 4 = 4 KiB outstanding per wave).
//
// NT: non-temporal stores in the 16B path — a bulk copy's destination is
// never re-read by this kernel, so dropping the lines from L2 leaves the
// cache to traffic that can reuse it (toggle measured on hardware).
// REMOTE: sources are IPC-mapped PEER memory (one-sided batched reads over
// xGMI, replacing per-piece SDMA enqueues); lines from a previous pull may
// be cached locally, so each workgroup issues one system-scope acquire
// before reading (validated by the cross-process staleness GPU test).
template <bool NT, bool REMOTE>
__global__ void __launch_bounds__(256)
copy_slices_kernel(const SliceDesc* __restrict__ descs, uint32_t nslices,
                   uint64_t total_units, uint32_t tile_bytes) {
  if (REMOTE) {
    if (threadIdx.x == 0) {
      __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
    }
    __syncthreads();
  }
  const uint32_t wave = threadIdx.x >> 6;
  const uint32_t lane = threadIdx.x & 63u;
  // BLOCK-CHUNKED assignment: each wave owns a CONTIGUOUS unit range, so
  // the slice descriptor is loaded once and advanced linearly — a strided
  // grid walk would re-search and re-load ~176 B of desc per unit, which
  // dominated small-row scatters (measured 249 GB/s in round 1)
  const uint64_t nwaves = (uint64_t)gridDim.x * 4;
  const uint64_t wave_id = (uint64_t)blockIdx.x * 4 + wave;
  const uint64_t chunk = (total_units + nwaves - 1) / nwaves;
  uint64_t unit = wave_id * chunk;
  const uint64_t unit_end =
      unit + chunk < total_units ? unit + chunk : total_units;
  if (unit >= unit_end) return;
  // binary search ONCE for the range start: greatest s with prefix <= unit
  uint32_t lo = 0, hi = nslices - 1;
  while (lo < hi) {
    uint32_t mid = (lo + hi + 1) >> 1;
    if (descs[mid].units_prefix <= unit) lo = mid; else hi = mid - 1;
  }
  SliceDesc d = descs[lo];
  uint64_t next_prefix = (lo + 1 < nslices)
                             ? descs[lo + 1].units_prefix
                             : ~0ull;
  for (; unit < unit_end; ++unit) {
    while (unit >= next_prefix) {
      ++lo;
      d = descs[lo];
      next_prefix = (lo + 1 < nslices) ? descs[lo + 1].units_prefix : ~0ull;
    }
    uint64_t local = unit - d.units_prefix;
    if (d.rows_per_unit > 1) {
      // row-packed unit: rows_per_unit consecutive rows, ndim<=1, all
      // 16B-aligned; a flat vectorized loop with div/mod row recovery
      // keeps ~4 independent 16B accesses in flight per lane
      uint64_t row0 = local * d.rows_per_unit;
      uint32_t nrows = (uint32_t)(d.rows - row0 < d.rows_per_unit
                                      ? d.rows - row0
                                      : d.rows_per_unit);
      int64_t sstride = d.ndim ? d.src_stride[0] : d.row_bytes;
      int64_t dstride = d.ndim ? d.dst_stride[0] : d.row_bytes;
      const char* sbase =
          reinterpret_cast<const char*>(d.src) + (int64_t)row0 * sstride;
      char* dbase = reinterpret_cast<char*>(d.dst) + (int64_t)row0 * dstride;
      const uint32_t rb4 = d.row_bytes >> 4;
      const uint32_t elems = nrows * rb4;
      uint32_t i = lane;
      if ((rb4 & (rb4 - 1)) == 0) {
        // power-of-two row: shift/mask replaces the per-element udiv pair
        // (~30 cycles each — visible in the LLC-resident regime)
        const uint32_t sh = 31 - __clz(rb4);
        const uint32_t msk = rb4 - 1;
        for (; i + 192 < elems; i += 256) {
          uint32_t i0 = i, i1 = i + 64, i2 = i + 128, i3 = i + 192;
          const uint4* s0 = reinterpret_cast<const uint4*>(
              sbase + (int64_t)(i0 >> sh) * sstride + ((i0 & msk) << 4));
          const uint4* s1 = reinterpret_cast<const uint4*>(
              sbase + (int64_t)(i1 >> sh) * sstride + ((i1 & msk) << 4));
          const uint4* s2 = reinterpret_cast<const uint4*>(
              sbase + (int64_t)(i2 >> sh) * sstride + ((i2 & msk) << 4));
          const uint4* s3 = reinterpret_cast<const uint4*>(
              sbase + (int64_t)(i3 >> sh) * sstride + ((i3 & msk) << 4));
          uint4 v0 = *s0;
          uint4 v1 = *s1;
          uint4 v2 = *s2;
          uint4 v3 = *s3;
          *reinterpret_cast<uint4*>(
              dbase + (int64_t)(i0 >> sh) * dstride + ((i0 & msk) << 4)) = v0;
          *reinterpret_cast<uint4*>(
              dbase + (int64_t)(i1 >> sh) * dstride + ((i1 & msk) << 4)) = v1;
          *reinterpret_cast<uint4*>(
              dbase + (int64_t)(i2 >> sh) * dstride + ((i2 & msk) << 4)) = v2;
          *reinterpret_cast<uint4*>(
              dbase + (int64_t)(i3 >> sh) * dstride + ((i3 & msk) << 4)) = v3;
        }
        for (; i < elems; i += 64) {
          *reinterpret_cast<uint4*>(
              dbase + (int64_t)(i >> sh) * dstride + ((i & msk) << 4)) =
              *reinterpret_cast<const uint4*>(
                  sbase + (int64_t)(i >> sh) * sstride + ((i & msk) << 4));
        }
        continue;
      }
      for (; i + 192 < elems; i += 256) {
        uint32_t i0 = i, i1 = i + 64, i2 = i + 128, i3 = i + 192;
        const uint4* s0 = reinterpret_cast<const uint4*>(
            sbase + (int64_t)(i0 / rb4) * sstride + ((i0 % rb4) << 4));
        const uint4* s1 = reinterpret_cast<const uint4*>(
            sbase + (int64_t)(i1 / rb4) * sstride + ((i1 % rb4) << 4));
        const uint4* s2 = reinterpret_cast<const uint4*>(
            sbase + (int64_t)(i2 / rb4) * sstride + ((i2 % rb4) << 4));
        const uint4* s3 = reinterpret_cast<const uint4*>(
            sbase + (int64_t)(i3 / rb4) * sstride + ((i3 % rb4) << 4));
        uint4 v0 = *s0;
        uint4 v1 = *s1;
        uint4 v2 = *s2;
        uint4 v3 = *s3;
        *reinterpret_cast<uint4*>(
            dbase + (int64_t)(i0 / rb4) * dstride + ((i0 % rb4) << 4)) = v0;
        *reinterpret_cast<uint4*>(
            dbase + (int64_t)(i1 / rb4) * dstride + ((i1 % rb4) << 4)) = v1;
        *reinterpret_cast<uint4*>(
            dbase + (int64_t)(i2 / rb4) * dstride + ((i2 % rb4) << 4)) = v2;
        *reinterpret_cast<uint4*>(
            dbase + (int64_t)(i3 / rb4) * dstride + ((i3 % rb4) << 4)) = v3;
      }
      for (; i < elems; i += 64) {
        *reinterpret_cast<uint4*>(
            dbase + (int64_t)(i / rb4) * dstride + ((i % rb4) << 4)) =
            *reinterpret_cast<const uint4*>(
                sbase + (int64_t)(i / rb4) * sstride + ((i % rb4) << 4));
      }
      continue;
    }
    uint64_t row = local / d.tiles_per_row;
    uint32_t tile = (uint32_t)(local % d.tiles_per_row);
    int64_t soff, doff;
    row_offsets(d, row, soff, doff);
    uint32_t start = tile * tile_bytes;
    uint32_t len = min(tile_bytes, d.row_bytes - start);
    const char* src = reinterpret_cast<const char*>(d.src) + soff + start;
    char* dst = reinterpret_cast<char*>(d.dst) + doff + start;
    uintptr_t sa = reinterpret_cast<uintptr_t>(src);
    uintptr_t da = reinterpret_cast<uintptr_t>(dst);
    if (((sa | da | len) & 15u) == 0) {
      const uint4* s4 = reinterpret_cast<const uint4*>(src);
      uint4* d4 = reinterpret_cast<uint4*>(dst);
      uint32_t n4 = len >> 4;
      uint32_t i = lane;
      // x8 unroll: 128 B of loads in flight per lane before the first
      // dependent store (HBM latency ~300 cyc wants deep MLP)
      for (; i + 448 < n4; i += 512) {
        uint4 v0 = s4[i];
        uint4 v1 = s4[i + 64];
        uint4 v2 = s4[i + 128];
        uint4 v3 = s4[i + 192];
        uint4 v4 = s4[i + 256];
        uint4 v5 = s4[i + 320];
        uint4 v6 = s4[i + 384];
        uint4 v7 = s4[i + 448];
        if (NT) {
          // uint4 is a class; nontemporal builtins need a real vector type
          typedef uint32_t v4u __attribute__((ext_vector_type(4)));
          v4u* dv = reinterpret_cast<v4u*>(d4);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v0), dv + i);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v1), dv + i + 64);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v2), dv + i + 128);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v3), dv + i + 192);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v4), dv + i + 256);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v5), dv + i + 320);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v6), dv + i + 384);
          __builtin_nontemporal_store(*reinterpret_cast<v4u*>(&v7), dv + i + 448);
        } else {
          d4[i] = v0;
          d4[i + 64] = v1;
          d4[i + 128] = v2;
          d4[i + 192] = v3;
          d4[i + 256] = v4;
          d4[i + 320] = v5;
          d4[i + 384] = v6;
          d4[i + 448] = v7;
        }
      }
      for (; i + 192 < n4; i += 256) {
        uint4 a = s4[i];
        uint4 b = s4[i + 64];
        uint4 c = s4[i + 128];
        uint4 e = s4[i + 192];
        d4[i] = a;
        d4[i + 64] = b;
        d4[i + 128] = c;
        d4[i + 192] = e;
      }
      for (; i < n4; i += 64) d4[i] = s4[i];
    } else if (((sa | da | len) & 3u) == 0) {
      const uint32_t* s1 = reinterpret_cast<const uint32_t*>(src);
      uint32_t* d1 = reinterpret_cast<uint32_t*>(dst);
      uint32_t n1 = len >> 2;
      for (uint32_t i = lane; i < n1; i += 64) d1[i] = s1[i];
    } else if (((sa | da | len) & 1u) == 0) {
      const uint16_t* s1 = reinterpret_cast<const uint16_t*>(src);
      uint16_t* d1 = reinterpret_cast<uint16_t*>(dst);
      uint32_t n1 = len >> 1;
      for (uint32_t i = lane; i < n1; i += 64) d1[i] = s1[i];
    } else {
      for (uint32_t i = lane; i < len; i += 64) dst[i] = src[i];
    }
  }
}

static bool use_nt_stores() {
  const char* e = getenv("HIPSTORE_NT");
  // default ON: +5% on 1 GiB flat copies, +26% on the reshard scatter
  // (profiles/gpu_primitive_microbench.log, NT1 vs NT0 runs)
  return e ? atoi(e) != 0 : true;
}

// upload descriptors through the pinned staging buffer + launch the kernel
static void launch_slice_descs(std::vector<SliceDesc>& descs, uint64_t units,
                               int device, hipStream_t stream,
                               uint32_t tile, bool remote = false) {
  if (units == 0 || descs.empty()) return;
  DevicePool& p = pool_for(device);
  HIP_CHECK(hipSetDevice(device));
  size_t bytes = descs.size() * sizeof(SliceDesc);
  std::lock_guard<std::mutex> lock(g_mutex);
  ensure_desc_capacity(p, device, bytes);
  if (p.desc_evt == nullptr) {
    HIP_CHECK(hipEventCreateWithFlags(&p.desc_evt, hipEventDisableTiming));
  } else {
    // the previous call's KERNEL must be done before h_desc/d_desc are
    // reused (calls may target different streams)
    HIP_CHECK(hipEventSynchronize(p.desc_evt));
  }
  std::memcpy(p.h_desc, descs.data(), bytes);
  HIP_CHECK(hipMemcpyAsync(p.d_desc, p.h_desc, bytes, hipMemcpyHostToDevice,
                           stream));
  // memory-bound: cap the grid and chunk units over it (guide G11);
  // each block consumes 4 contiguous unit ranges (one per wave)
  uint32_t grid = (uint32_t)std::min<uint64_t>((units + 3) / 4, pick_grid_cap());
  auto kern = use_nt_stores()
                  ? (remote ? copy_slices_kernel<true, true>
                            : copy_slices_kernel<true, false>)
                  : (remote ? copy_slices_kernel<false, true>
                            : copy_slices_kernel<false, false>);
  hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<const SliceDesc*>(p.d_desc),
                     (uint32_t)descs.size(), units, tile);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipEventRecord(p.desc_evt, stream));
}

// trivial desc for a flat byte copy (used by copy_batch's same-device path)
static SliceDesc flat_desc(uintptr_t dst, uintptr_t src, uint64_t nbytes,
                           uint64_t units_prefix, uint32_t tile) {
  SliceDesc d{};
  d.src = src;
  d.dst = dst;
  d.rows = 1;
  d.units_prefix = units_prefix;
  d.row_bytes = (uint32_t)nbytes;
  d.tiles_per_row = (uint32_t)((nbytes + tile - 1) / tile);
  if (d.tiles_per_row == 0) d.tiles_per_row = 1;
  d.ndim = 0;
  return d;
}

static void copy_batch(
    const std::vector<std::tuple<uintptr_t, int, uintptr_t, int, uint64_t>>&
        copies) {
  if (copies.empty()) return;
  // group same-device copies per device; cross-device go via SDMA
  std::unordered_map<int, std::vector<std::tuple<uintptr_t, uintptr_t, uint64_t>>>
      flat_per_device;
  std::vector<hipStream_t> used;
  int i = 0;
  for (const auto& c : copies) {
    uintptr_t dst = std::get<0>(c);
    int dst_dev = std::get<1>(c);
    uintptr_t src = std::get<2>(c);
    int src_dev = std::get<3>(c);
    uint64_t n = std::get<4>(c);
    if (n == 0) continue;
    if (dst_dev == src_dev && n <= UINT32_MAX) {
      flat_per_device[dst_dev].emplace_back(dst, src, n);
      continue;
    }
    DevicePool& p = pool_for(dst_dev);
    hipStream_t s = p.streams[i++ % kStreamsPerDevice];
    HIP_CHECK(hipSetDevice(dst_dev));
    // UVA-resolved copy: the canonical form for IPC-mapped peer pointers
    // (hipMemcpyPeerAsync requires context-owned pointers on both ends;
    // Default kind lets the driver resolve mapped peer memory)
    HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dst),
                             reinterpret_cast<void*>(src), n,
                             hipMemcpyDefault, s));
    (void)src_dev;
    used.push_back(s);
  }
  for (auto& kv : flat_per_device) {
    int device = kv.first;
    uint64_t total = 0;
    for (auto& t : kv.second) total += std::get<2>(t);
    uint32_t tile = pick_tile(total);
    std::vector<SliceDesc> descs;
    descs.reserve(kv.second.size());
    uint64_t units = 0;
    for (auto& t : kv.second) {
      descs.push_back(flat_desc(std::get<0>(t), std::get<1>(t),
                                std::get<2>(t), units, tile));
      units += descs.back().tiles_per_row;
    }
    DevicePool& p = pool_for(device);
    hipStream_t s = p.streams[0];
    launch_slice_descs(descs, units, device, s, tile);
    used.push_back(s);
  }
  for (hipStream_t s : used) HIP_CHECK(hipStreamSynchronize(s));
}

// python passes per slice:
//   (src_ptr, dst_ptr, row_bytes, [outer shape], [src strides B], [dst strides B])
using PySlice = std::tuple<uintptr_t, uintptr_t, uint64_t,
                           std::vector<uint64_t>, std::vector<int64_t>,
                           std::vector<int64_t>>;

static void copy_slices(const std::vector<PySlice>& slices, int device,
                        uintptr_t stream_handle, bool blocking,
                        bool remote) {
  if (slices.empty()) return;
  size_t n = slices.size();
  uint64_t total_bytes = 0;
  for (const auto& s : slices) {
    uint64_t rows = 1;
    for (uint64_t d : std::get<3>(s)) rows *= d;
    total_bytes += rows * std::get<2>(s);
  }
  uint32_t tile = pick_tile(total_bytes);
  std::vector<SliceDesc> descs(n);
  uint64_t units = 0;
  for (size_t i = 0; i < n; ++i) {
    const auto& s = slices[i];
    SliceDesc& d = descs[i];
    d.src = std::get<0>(s);
    d.dst = std::get<1>(s);
    uint64_t row_bytes = std::get<2>(s);
    const auto& shape = std::get<3>(s);
    const auto& sst = std::get<4>(s);
    const auto& dst = std::get<5>(s);
    if (shape.size() > kMaxDims) throw std::runtime_error("too many dims");
    if (row_bytes > UINT32_MAX) throw std::runtime_error("row too large");
    d.ndim = (uint32_t)shape.size();
    d.rows = 1;
    for (size_t k = 0; k < shape.size(); ++k) {
      d.shape[k] = shape[k];
      d.src_stride[k] = sst[k];
      d.dst_stride[k] = dst[k];
      d.rows *= shape[k];
    }
    d.row_bytes = (uint32_t)row_bytes;
    d.tiles_per_row = (uint32_t)((row_bytes + tile - 1) / tile);
    if (d.tiles_per_row == 0) d.tiles_per_row = 1;
    d.rows_per_unit = 1;
    // row-packed eligibility: small fully-16B-aligned rows, <=1 outer dim
    bool aligned16 = row_bytes > 0 && (row_bytes & 15u) == 0 &&
                     (d.src & 15u) == 0 && (d.dst & 15u) == 0 &&
                     (d.ndim == 0 ||
                      ((d.src_stride[0] & 15) == 0 &&
                       (d.dst_stride[0] & 15) == 0));
    if (d.ndim <= 1 && d.rows > 1 && row_bytes < kRowPackSpan && aligned16) {
      d.rows_per_unit = (uint32_t)(kRowPackSpan / row_bytes);
      d.tiles_per_row = 1;
    }
    d.units_prefix = units;
    if (d.rows_per_unit > 1) {
      units += (d.rows + d.rows_per_unit - 1) / d.rows_per_unit;
    } else {
      units += d.rows * d.tiles_per_row;
    }
  }
  if (units == 0) return;

  launch_slice_descs(descs, units, device,
                     reinterpret_cast<hipStream_t>(stream_handle), tile,
                     remote);
  if (blocking) {
    HIP_CHECK(
        hipStreamSynchronize(reinterpret_cast<hipStream_t>(stream_handle)));
  }
}

// ---------------------------------------------------------------------------
// K3 — fused cast + pack
// ---------------------------------------------------------------------------

enum class DType : int32_t { F32 = 0, F16 = 1, BF16 = 2 };

template <typename SrcT, typename DstT>
__device__ __forceinline__ DstT convert(SrcT v);

template <> __device__ __forceinline__ float convert<float, float>(float v) { return v; }
template <> __device__ __forceinline__ __hip_bfloat16 convert<float, __hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <> __device__ __forceinline__ float convert<__hip_bfloat16, float>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <> __device__ __forceinline__ __half convert<float, __half>(float v) {
  return __float2half(v);
}
template <> __device__ __forceinline__ float convert<__half, float>(__half v) {
  return __half2float(v);
}
template <> __device__ __forceinline__ __half convert<__hip_bfloat16, __half>(__hip_bfloat16 v) {
  return __float2half(__bfloat162float(v));
}
template <> __device__ __forceinline__ __hip_bfloat16 convert<__half, __hip_bfloat16>(__half v) {
  return __float2bfloat16(__half2float(v));
}

// V elements per thread per iteration; 16B loads when SrcT is 2 bytes,
// 32B (2x float4) when 4 bytes at V=8 — always >= the 16B/lane coalescing
// sweet spot on the wider side (guide G13).  HIPSTORE_CAST_V sweeps V.
template <typename SrcT, typename DstT, uint32_t V>
__global__ void __launch_bounds__(256)
cast_copy_kernel(const SrcT* __restrict__ src, DstT* __restrict__ dst,
                 uint64_t numel) {
  uint64_t base = (uint64_t)(blockIdx.x * blockDim.x + threadIdx.x) * V;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x * V;
  struct alignas(16) SrcVec { SrcT v[V]; };
  struct alignas(16) DstVec { DstT v[V]; };
  for (uint64_t i = base; i + V <= numel; i += stride) {
    SrcVec s = *reinterpret_cast<const SrcVec*>(src + i);
    DstVec d;
#pragma unroll
    for (uint32_t k = 0; k < V; ++k) d.v[k] = convert<SrcT, DstT>(s.v[k]);
    *reinterpret_cast<DstVec*>(dst + i) = d;
  }
  // tail
  uint64_t tail_start = (numel / V) * V;
  uint64_t tid = blockIdx.x * blockDim.x + threadIdx.x;
  if (tid < numel - tail_start) {
    uint64_t i = tail_start + tid;
    dst[i] = convert<SrcT, DstT>(src[i]);
  }
}

template <typename SrcT, typename DstT>
static void launch_cast(uintptr_t src, uintptr_t dst, uint64_t numel,
                        hipStream_t stream) {
  const char* e = getenv("HIPSTORE_CAST_V");
  uint32_t v = e ? (uint32_t)atoi(e) : 8u;
  uint64_t work = (numel + v - 1) / v;
  uint32_t grid =
      (uint32_t)std::min<uint64_t>((work + 255) / 256, pick_cast_grid_cap());
  if (grid == 0) grid = 1;
  if (v >= 16) {
    hipLaunchKernelGGL((cast_copy_kernel<SrcT, DstT, 16>), dim3(grid),
                       dim3(256), 0, stream,
                       reinterpret_cast<const SrcT*>(src),
                       reinterpret_cast<DstT*>(dst), numel);
  } else {
    hipLaunchKernelGGL((cast_copy_kernel<SrcT, DstT, 8>), dim3(grid),
                       dim3(256), 0, stream,
                       reinterpret_cast<const SrcT*>(src),
                       reinterpret_cast<DstT*>(dst), numel);
  }
  HIP_CHECK(hipGetLastError());
}

static void cast_copy(uintptr_t src, int src_dtype, uintptr_t dst,
                      int dst_dtype, uint64_t numel, int device,
                      uintptr_t stream_handle) {
  HIP_CHECK(hipSetDevice(device));
  hipStream_t stream = reinterpret_cast<hipStream_t>(stream_handle);
  auto s = static_cast<DType>(src_dtype);
  auto d = static_cast<DType>(dst_dtype);
  if (s == DType::F32 && d == DType::BF16)
    launch_cast<float, __hip_bfloat16>(src, dst, numel, stream);
  else if (s == DType::BF16 && d == DType::F32)
    launch_cast<__hip_bfloat16, float>(src, dst, numel, stream);
  else if (s == DType::F32 && d == DType::F16)
    launch_cast<float, __half>(src, dst, numel, stream);
  else if (s == DType::F16 && d == DType::F32)
    launch_cast<__half, float>(src, dst, numel, stream);
  else if (s == DType::BF16 && d == DType::F16)
    launch_cast<__hip_bfloat16, __half>(src, dst, numel, stream);
  else if (s == DType::F16 && d == DType::BF16)
    launch_cast<__half, __hip_bfloat16>(src, dst, numel, stream);
  else
    throw std::runtime_error("unsupported cast pair");
}

// ---------------------------------------------------------------------------
// host pinning
// ---------------------------------------------------------------------------

// single UVA-resolved copy on a dedicated pool stream (blocking).  Used by
// the SHM transport for D2H staging: torch's copy_ into externally
// registered host memory runs ~2x slower than a direct hipMemcpyAsync.
static void sdma_copy(uintptr_t dst, uintptr_t src, uint64_t nbytes,
                      int device) {
  DevicePool& p = pool_for(device);
  HIP_CHECK(hipSetDevice(device));
  hipStream_t s = p.streams[1];
  HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dst),
                           reinterpret_cast<void*>(src), nbytes,
                           hipMemcpyDefault, s));
  HIP_CHECK(hipStreamSynchronize(s));
}

static void host_register(uintptr_t ptr, uint64_t nbytes) {
  HIP_CHECK(hipHostRegister(reinterpret_cast<void*>(ptr), nbytes,
                            hipHostRegisterPortable));
}

static void host_unregister(uintptr_t ptr) {
  HIP_CHECK(hipHostUnregister(reinterpret_cast<void*>(ptr)));
}

static int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  return e == hipSuccess ? n : 0;
}

static void sync_device(int device) {
  HIP_CHECK(hipSetDevice(device));
  HIP_CHECK(hipDeviceSynchronize());
}

PYBIND11_MODULE(_hipstore, m) {
  m.doc() = "torchstore_amd native core (HIP/CDNA4, gfx950)";
  m.def("ipc_export", &ipc_export, py::arg("ptr"), py::arg("device"),
        py::arg("generation"));
  m.def("ipc_export_cache_clear", &ipc_export_cache_clear);
  m.def("ipc_open", &ipc_open, py::arg("handle"), py::arg("local_device"),
        py::arg("src_device"));
  m.def("ipc_close", &ipc_close, py::arg("base"), py::arg("local_device"));
  m.def("copy_batch", &copy_batch, py::arg("copies"),
        py::call_guard<py::gil_scoped_release>());
  m.def("copy_batch_2d", &copy_batch_2d, py::arg("copies"),
        py::call_guard<py::gil_scoped_release>());
  m.def("copy_slices", &copy_slices, py::arg("slices"), py::arg("device"),
        py::arg("stream"), py::arg("blocking") = true,
        py::arg("remote") = false,
        py::call_guard<py::gil_scoped_release>());
  m.def("cast_copy", &cast_copy, py::arg("src"), py::arg("src_dtype"),
        py::arg("dst"), py::arg("dst_dtype"), py::arg("numel"),
        py::arg("device"), py::arg("stream"));
  m.def("sdma_copy", &sdma_copy, py::arg("dst"), py::arg("src"),
        py::arg("nbytes"), py::arg("device"),
        py::call_guard<py::gil_scoped_release>());
  m.def("host_register", &host_register);
  m.def("host_unregister", &host_unregister);
  m.def("device_count", &device_count);
  m.def("sync_device", &sync_device);
}
