// Fused SGNS training kernel for MI355X (gfx950, CDNA4).
//
// This is the MI355X-native replacement for the reference's entire hot path:
// the per-minibatch client loop (mllib ServerSideGlintWord2Vec.scala:419-429)
// plus the Glint parameter-server dotprod/adjust ops.  One
// kernel launch consumes a step's worth of encoded sentences and performs,
// fully fused on-device:
//   frequency subsampling -> shrunk-window pair generation -> unigram-table
//   negative draws -> embedding row gather -> center*target dots (wave
//   reduction) -> clipped sigmoid + gradient -> SGD scatter-add on both
//   tables.
//
// Execution model: one wave per sentence (4 waves per 256-thread
// workgroup), sentences round-robin over the launched waves; long
// sentences additionally split into position blocks over gridDim.y.  The
// shipped variant runs TWO pairs per wave (32-lane halves — sgns_train2),
// with 64-lane (exact oracle order, used for serial parity) and 16-lane
// four-pair variants selectable via pair_mode.  The per-GPU asynchrony of
// the reference's numPartitions workers (mllib:120-127) becomes hogwild
// waves: row updates are plain read-modify-write (the reference's
// fire-and-forget adjust), with atomic modes (all rows / hot rows /
// positive pairs) for the measured quality/speed frontier
// (benchmarks/results.md).
//
// Rows are laid out [vocab][stride] with stride a multiple of 64 elements;
// each lane owns elements in chunk-pairs of 128 (2 adjacent elements per
// lane -> dwordx2/dword coalesced 512B/256B per instruction, guide G13),
// plus one 64-element tail chunk when NC is odd.  Padding elements stay 0
// forever (0-init, updates scale by row values), so no masking is needed.
//
// RNG: counter-based splitmix64 streams, bit-identical to
// glint_word2vec_amd/rng.py and csrc/cpu_sgns.cpp — see rng.py for the
// normative draw-index layout.  This makes every draw computable at any
// lane with no sequential state.

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>

namespace py = pybind11;

// hipGetLastError is sticky per-thread: clear stale errors (e.g. from
// torch's own device probing) at wrapper entry so HIP_CHECK after our
// launches reports only our result.
#define HIP_CLEAR_ERROR() (void)hipGetLastError()
#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess)                                                   \
      throw std::runtime_error(std::string("HIP error: ") +                 \
                               hipGetErrorString(_e) + " at " __FILE__ ":" + \
                               std::to_string(__LINE__));                   \
  } while (0)

// ---------------------------------------------------------------------------
// RNG (normative contract in glint_word2vec_amd/rng.py)
// ---------------------------------------------------------------------------
__host__ __device__ __forceinline__ uint64_t splitmix64(uint64_t x) {
  uint64_t z = x + 0x9E3779B97F4A7C15ULL;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}
__host__ __device__ __forceinline__ uint64_t sent_base(uint64_t seed, uint64_t sid) {
  return splitmix64(seed ^ (sid * 0x9E3779B97F4A7C15ULL));
}
__device__ __forceinline__ uint32_t draw_u32(uint64_t base, uint64_t k) {
  return (uint32_t)(splitmix64(base + k * 0x9E3779B97F4A7C15ULL) >> 32);
}
constexpr uint64_t kWinBase = 1ULL << 20;
constexpr uint64_t kNegBase = 1ULL << 21;

// ---------------------------------------------------------------------------
// Row I/O: float32 and bf16 (raw u16) storage, f32 math.
// Element of register slot k (k < NC, NC = stride/64):
//   paired slots 2m / 2m+1 : elements 128*m + 2*lane + {0,1}
//   odd-NC tail slot NC-1  : element  64*(NC-1) + lane
// ---------------------------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) __bf16 v2bf16;

__device__ __forceinline__ float bf16_to_f32(uint16_t h) {
  uint32_t u = ((uint32_t)h) << 16;
  return __uint_as_float(u);
}
__device__ __forceinline__ uint16_t f32_to_bf16_rne(float f) {
  uint32_t x = __float_as_uint(f);
  uint32_t r = (x + 0x7FFFu + ((x >> 16) & 1u)) >> 16;
  return (uint16_t)r;
}

template <typename T, int NC>
struct RowIO;

template <int NC>
struct RowIO<float, NC> {
  static __device__ __forceinline__ void load(const float* row, float v[NC], int lane) {
#pragma unroll
    for (int m = 0; m < NC / 2; ++m) {
      const float2 p = *reinterpret_cast<const float2*>(row + 128 * m + 2 * lane);
      v[2 * m] = p.x;
      v[2 * m + 1] = p.y;
    }
    if (NC & 1) v[NC - 1] = row[64 * (NC - 1) + lane];
  }
  static __device__ __forceinline__ void store(float* row, const float v[NC], int lane) {
#pragma unroll
    for (int m = 0; m < NC / 2; ++m) {
      *reinterpret_cast<float2*>(row + 128 * m + 2 * lane) =
          make_float2(v[2 * m], v[2 * m + 1]);
    }
    if (NC & 1) row[64 * (NC - 1) + lane] = v[NC - 1];
  }
  static __device__ __forceinline__ void atomic_add(float* row, const float v[NC], int lane) {
#pragma unroll
    for (int m = 0; m < NC / 2; ++m) {
      atomicAdd(row + 128 * m + 2 * lane, v[2 * m]);
      atomicAdd(row + 128 * m + 2 * lane + 1, v[2 * m + 1]);
    }
    if (NC & 1) atomicAdd(row + 64 * (NC - 1) + lane, v[NC - 1]);
  }
};

template <int NC>
struct RowIO<uint16_t, NC> {
  static __device__ __forceinline__ void load(const uint16_t* row, float v[NC], int lane) {
#pragma unroll
    for (int m = 0; m < NC / 2; ++m) {
      const uint32_t p = *reinterpret_cast<const uint32_t*>(row + 128 * m + 2 * lane);
      v[2 * m] = bf16_to_f32((uint16_t)(p & 0xFFFF));
      v[2 * m + 1] = bf16_to_f32((uint16_t)(p >> 16));
    }
    if (NC & 1) v[NC - 1] = bf16_to_f32(row[64 * (NC - 1) + lane]);
  }
  static __device__ __forceinline__ void store(uint16_t* row, const float v[NC], int lane) {
    // plain casts: hipcc fuses pairs into v_cvt_pk_bf16_f32 (RNE), one
    // instruction per two elements vs ~4 VALU each for bit-math RNE
#pragma unroll
    for (int m = 0; m < NC / 2; ++m) {
      v2bf16 d;
      d[0] = (__bf16)v[2 * m];
      d[1] = (__bf16)v[2 * m + 1];
      *reinterpret_cast<v2bf16*>(row + 128 * m + 2 * lane) = d;
    }
    if (NC & 1)
      *reinterpret_cast<__bf16*>(row + 64 * (NC - 1) + lane) = (__bf16)v[NC - 1];
  }
  // gfx950 packed-bf16 atomic add (global_atomic_pk_add_bf16): adds the f32
  // delta rounded to bf16 — no lost updates on contended (hot Zipf) rows.
  static __device__ __forceinline__ void atomic_add(uint16_t* row, const float v[NC], int lane) {
#pragma unroll
    for (int m = 0; m < NC / 2; ++m) {
      v2bf16 d;
      d[0] = (__bf16)v[2 * m];
      d[1] = (__bf16)v[2 * m + 1];
      __builtin_amdgcn_global_atomic_fadd_v2bf16(
          (v2bf16*)(row + 128 * m + 2 * lane), d);
    }
    if (NC & 1) {
      // tail: one element per lane; hit the containing aligned pair with the
      // other half = +0.0 (x + 0.0 == x in bf16)
      uint16_t* p = row + 64 * (NC - 1) + lane;
      const bool odd = ((uintptr_t)p >> 1) & 1;
      v2bf16 d;
      d[0] = odd ? (__bf16)0.0f : (__bf16)v[NC - 1];
      d[1] = odd ? (__bf16)v[NC - 1] : (__bf16)0.0f;
      __builtin_amdgcn_global_atomic_fadd_v2bf16(
          (v2bf16*)((uintptr_t)p & ~(uintptr_t)3), d);
    }
  }
};

// ---------------------------------------------------------------------------
// 32-lane row I/O: each HALF-wave owns one row (two pairs per wave — 2x the
// rows in flight per wave at the same instruction count).  Element of slot
// m (m < NCH, NCH = stride/32): pairs of slots 2q/2q+1 = elements
// 64*q + 2*(lane&31) + {0,1}; odd-NCH tail slot = 32*(NCH-1) + (lane&31).
// ---------------------------------------------------------------------------
template <typename T, int NCH>
struct RowIO32;

template <int NCH>
struct RowIO32<float, NCH> {
  static __device__ __forceinline__ void load(const float* row, float v[NCH], int l32) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const float2 p = *reinterpret_cast<const float2*>(row + 64 * q + 2 * l32);
      v[2 * q] = p.x;
      v[2 * q + 1] = p.y;
    }
    if (NCH & 1) v[NCH - 1] = row[32 * (NCH - 1) + l32];
  }
  static __device__ __forceinline__ void store(float* row, const float v[NCH], int l32) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q)
      *reinterpret_cast<float2*>(row + 64 * q + 2 * l32) =
          make_float2(v[2 * q], v[2 * q + 1]);
    if (NCH & 1) row[32 * (NCH - 1) + l32] = v[NCH - 1];
  }
  static __device__ __forceinline__ void atomic_add(float* row, const float v[NCH], int l32) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      atomicAdd(row + 64 * q + 2 * l32, v[2 * q]);
      atomicAdd(row + 64 * q + 2 * l32 + 1, v[2 * q + 1]);
    }
    if (NCH & 1) atomicAdd(row + 32 * (NCH - 1) + l32, v[NCH - 1]);
  }
};

template <int NCH>
struct RowIO32<uint16_t, NCH> {
  static __device__ __forceinline__ void load(const uint16_t* row, float v[NCH], int l32) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const uint32_t p = *reinterpret_cast<const uint32_t*>(row + 64 * q + 2 * l32);
      v[2 * q] = bf16_to_f32((uint16_t)(p & 0xFFFF));
      v[2 * q + 1] = bf16_to_f32((uint16_t)(p >> 16));
    }
    if (NCH & 1) v[NCH - 1] = bf16_to_f32(row[32 * (NCH - 1) + l32]);
  }
  static __device__ __forceinline__ void store(uint16_t* row, const float v[NCH], int l32) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      v2bf16 d;
      d[0] = (__bf16)v[2 * q];
      d[1] = (__bf16)v[2 * q + 1];
      *reinterpret_cast<v2bf16*>(row + 64 * q + 2 * l32) = d;
    }
    if (NCH & 1)
      *reinterpret_cast<__bf16*>(row + 32 * (NCH - 1) + l32) = (__bf16)v[NCH - 1];
  }
  static __device__ __forceinline__ void atomic_add(uint16_t* row, const float v[NCH], int l32) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      v2bf16 d;
      d[0] = (__bf16)v[2 * q];
      d[1] = (__bf16)v[2 * q + 1];
      __builtin_amdgcn_global_atomic_fadd_v2bf16(
          (v2bf16*)(row + 64 * q + 2 * l32), d);
    }
    if (NCH & 1) {
      uint16_t* p = row + 32 * (NCH - 1) + l32;
      const bool odd = ((uintptr_t)p >> 1) & 1;
      v2bf16 d;
      d[0] = odd ? (__bf16)0.0f : (__bf16)v[NCH - 1];
      d[1] = odd ? (__bf16)v[NCH - 1] : (__bf16)0.0f;
      __builtin_amdgcn_global_atomic_fadd_v2bf16(
          (v2bf16*)((uintptr_t)p & ~(uintptr_t)3), d);
    }
  }
};

// 16-lane row I/O (four pairs per wave).  Element of slot m (m < NCQ,
// NCQ = stride/16): pairs of slots 2q/2q+1 = elements 32*q + 2*(lane&15)
// + {0,1}; odd tail slot = 16*(NCQ-1) + (lane&15).
template <typename T, int NCQ>
struct RowIO16;

template <int NCQ>
struct RowIO16<float, NCQ> {
  static __device__ __forceinline__ void load(const float* row, float v[NCQ], int l16) {
#pragma unroll
    for (int q = 0; q < NCQ / 2; ++q) {
      const float2 p = *reinterpret_cast<const float2*>(row + 32 * q + 2 * l16);
      v[2 * q] = p.x;
      v[2 * q + 1] = p.y;
    }
    if (NCQ & 1) v[NCQ - 1] = row[16 * (NCQ - 1) + l16];
  }
  static __device__ __forceinline__ void store(float* row, const float v[NCQ], int l16) {
#pragma unroll
    for (int q = 0; q < NCQ / 2; ++q)
      *reinterpret_cast<float2*>(row + 32 * q + 2 * l16) =
          make_float2(v[2 * q], v[2 * q + 1]);
    if (NCQ & 1) row[16 * (NCQ - 1) + l16] = v[NCQ - 1];
  }
  static __device__ __forceinline__ void atomic_add(float* row, const float v[NCQ], int l16) {
#pragma unroll
    for (int q = 0; q < NCQ / 2; ++q) {
      atomicAdd(row + 32 * q + 2 * l16, v[2 * q]);
      atomicAdd(row + 32 * q + 2 * l16 + 1, v[2 * q + 1]);
    }
    if (NCQ & 1) atomicAdd(row + 16 * (NCQ - 1) + l16, v[NCQ - 1]);
  }
};

template <int NCQ>
struct RowIO16<uint16_t, NCQ> {
  static __device__ __forceinline__ void load(const uint16_t* row, float v[NCQ], int l16) {
#pragma unroll
    for (int q = 0; q < NCQ / 2; ++q) {
      const uint32_t p = *reinterpret_cast<const uint32_t*>(row + 32 * q + 2 * l16);
      v[2 * q] = bf16_to_f32((uint16_t)(p & 0xFFFF));
      v[2 * q + 1] = bf16_to_f32((uint16_t)(p >> 16));
    }
    if (NCQ & 1) v[NCQ - 1] = bf16_to_f32(row[16 * (NCQ - 1) + l16]);
  }
  static __device__ __forceinline__ void store(uint16_t* row, const float v[NCQ], int l16) {
#pragma unroll
    for (int q = 0; q < NCQ / 2; ++q) {
      v2bf16 d;
      d[0] = (__bf16)v[2 * q];
      d[1] = (__bf16)v[2 * q + 1];
      *reinterpret_cast<v2bf16*>(row + 32 * q + 2 * l16) = d;
    }
    if (NCQ & 1)
      *reinterpret_cast<__bf16*>(row + 16 * (NCQ - 1) + l16) = (__bf16)v[NCQ - 1];
  }
  static __device__ __forceinline__ void atomic_add(uint16_t* row, const float v[NCQ], int l16) {
#pragma unroll
    for (int q = 0; q < NCQ / 2; ++q) {
      v2bf16 d;
      d[0] = (__bf16)v[2 * q];
      d[1] = (__bf16)v[2 * q + 1];
      __builtin_amdgcn_global_atomic_fadd_v2bf16(
          (v2bf16*)(row + 32 * q + 2 * l16), d);
    }
    if (NCQ & 1) {
      uint16_t* p = row + 16 * (NCQ - 1) + l16;
      const bool odd = ((uintptr_t)p >> 1) & 1;
      v2bf16 d;
      d[0] = odd ? (__bf16)0.0f : (__bf16)v[NCQ - 1];
      d[1] = odd ? (__bf16)v[NCQ - 1] : (__bf16)0.0f;
      __builtin_amdgcn_global_atomic_fadd_v2bf16(
          (v2bf16*)((uintptr_t)p & ~(uintptr_t)3), d);
    }
  }
};

// Per-quarter (16-lane) sum: row_shr prefix leaves each 16-row's total in
// its lane 15; width-16 shuffle broadcasts it within the quarter.
__device__ __forceinline__ float quarter_sum_f32(float v) {
  typedef int i32;
  i32 x = __float_as_int(v);
#define DPPQ_ADD(ctrl)                                                       \
  x = __float_as_int(__int_as_float(x) +                                     \
      __int_as_float(__builtin_amdgcn_update_dpp(0, x, ctrl, 0xF, 0xF, true)))
  DPPQ_ADD(0x111);
  DPPQ_ADD(0x112);
  DPPQ_ADD(0x114);
  DPPQ_ADD(0x118);
#undef DPPQ_ADD
  return __shfl(__int_as_float(x), 15, 16);
}

// Masked 32-lane row I/O for dim-slice tables whose width is not a
// multiple of 64 (dim-sharded slices, e.g. dim 300 over 8 GPUs = 38 wide):
// rows are stored at stride = round_up(width, 8) elements and lanes beyond
// the width contribute zeros.  Only the dim-sharded phases pay the guard
// cost; the fused train kernels keep the unmasked RowIO32.
template <typename T, int NCH>
struct RowIO32M;

template <int NCH>
struct RowIO32M<float, NCH> {
  static __device__ __forceinline__ void load(const float* row, float v[NCH],
                                              int l32, int width) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const int idx = 64 * q + 2 * l32;
      if (idx + 1 < width) {
        const float2 p = *reinterpret_cast<const float2*>(row + idx);
        v[2 * q] = p.x;
        v[2 * q + 1] = p.y;
      } else if (idx < width) {
        v[2 * q] = row[idx];
        v[2 * q + 1] = 0.0f;
      } else {
        v[2 * q] = 0.0f;
        v[2 * q + 1] = 0.0f;
      }
    }
    if (NCH & 1) {
      const int idx = 32 * (NCH - 1) + l32;
      v[NCH - 1] = idx < width ? row[idx] : 0.0f;
    }
  }
  static __device__ __forceinline__ void store(float* row, const float v[NCH],
                                               int l32, int width) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const int idx = 64 * q + 2 * l32;
      if (idx + 1 < width) {
        *reinterpret_cast<float2*>(row + idx) =
            make_float2(v[2 * q], v[2 * q + 1]);
      } else if (idx < width) {
        row[idx] = v[2 * q];
      }
    }
    if (NCH & 1) {
      const int idx = 32 * (NCH - 1) + l32;
      if (idx < width) row[idx] = v[NCH - 1];
    }
  }
  static __device__ __forceinline__ void atomic_add(float* row,
                                                    const float v[NCH],
                                                    int l32, int width) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const int idx = 64 * q + 2 * l32;
      if (idx < width) atomicAdd(row + idx, v[2 * q]);
      if (idx + 1 < width) atomicAdd(row + idx + 1, v[2 * q + 1]);
    }
    if (NCH & 1) {
      const int idx = 32 * (NCH - 1) + l32;
      if (idx < width) atomicAdd(row + idx, v[NCH - 1]);
    }
  }
};

template <int NCH>
struct RowIO32M<uint16_t, NCH> {
  static __device__ __forceinline__ void load(const uint16_t* row, float v[NCH],
                                              int l32, int width) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const int idx = 64 * q + 2 * l32;
      if (idx + 1 < width) {
        const uint32_t p = *reinterpret_cast<const uint32_t*>(row + idx);
        v[2 * q] = bf16_to_f32((uint16_t)(p & 0xFFFF));
        v[2 * q + 1] = bf16_to_f32((uint16_t)(p >> 16));
      } else if (idx < width) {
        v[2 * q] = bf16_to_f32(row[idx]);
        v[2 * q + 1] = 0.0f;
      } else {
        v[2 * q] = 0.0f;
        v[2 * q + 1] = 0.0f;
      }
    }
    if (NCH & 1) {
      const int idx = 32 * (NCH - 1) + l32;
      v[NCH - 1] = idx < width ? bf16_to_f32(row[idx]) : 0.0f;
    }
  }
  static __device__ __forceinline__ void store(uint16_t* row, const float v[NCH],
                                               int l32, int width) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const int idx = 64 * q + 2 * l32;
      if (idx + 1 < width) {
        v2bf16 d;
        d[0] = (__bf16)v[2 * q];
        d[1] = (__bf16)v[2 * q + 1];
        *reinterpret_cast<v2bf16*>(row + idx) = d;
      } else if (idx < width) {
        *reinterpret_cast<__bf16*>(row + idx) = (__bf16)v[2 * q];
      }
    }
    if (NCH & 1) {
      const int idx = 32 * (NCH - 1) + l32;
      if (idx < width) *reinterpret_cast<__bf16*>(row + idx) = (__bf16)v[NCH - 1];
    }
  }
  static __device__ __forceinline__ void atomic_add(uint16_t* row,
                                                    const float v[NCH],
                                                    int l32, int width) {
#pragma unroll
    for (int q = 0; q < NCH / 2; ++q) {
      const int idx = 64 * q + 2 * l32;
      if (idx + 1 < width) {
        v2bf16 d;
        d[0] = (__bf16)v[2 * q];
        d[1] = (__bf16)v[2 * q + 1];
        __builtin_amdgcn_global_atomic_fadd_v2bf16((v2bf16*)(row + idx), d);
      } else if (idx < width) {
        // lone element at an even offset: pack with a zero partner
        v2bf16 d;
        d[0] = (__bf16)v[2 * q];
        d[1] = (__bf16)0.0f;
        __builtin_amdgcn_global_atomic_fadd_v2bf16((v2bf16*)(row + idx), d);
      }
    }
    if (NCH & 1) {
      const int idx = 32 * (NCH - 1) + l32;
      if (idx < width) {
        uint16_t* p = row + idx;
        const bool odd = ((uintptr_t)p >> 1) & 1;
        v2bf16 d;
        d[0] = odd ? (__bf16)0.0f : (__bf16)v[NCH - 1];
        d[1] = odd ? (__bf16)v[NCH - 1] : (__bf16)0.0f;
        __builtin_amdgcn_global_atomic_fadd_v2bf16(
            (v2bf16*)((uintptr_t)p & ~(uintptr_t)3), d);
      }
    }
  }
};

// Compile-time masked/unmasked row-I/O selector: padded storage keeps the
// original unmasked vector loads (the masked variant's per-chunk guards
// cost real throughput even when always-true); narrow storage masks.
template <typename T, int NCH, bool MASKED>
struct RIO32 {
  static __device__ __forceinline__ void load(const T* row, float v[NCH],
                                              int l32, int width) {
    if (MASKED) RowIO32M<T, NCH>::load(row, v, l32, width);
    else RowIO32<T, NCH>::load(row, v, l32);
  }
  static __device__ __forceinline__ void store(T* row, const float v[NCH],
                                               int l32, int width) {
    if (MASKED) RowIO32M<T, NCH>::store(row, v, l32, width);
    else RowIO32<T, NCH>::store(row, v, l32);
  }
  static __device__ __forceinline__ void atomic_add(T* row,
                                                    const float v[NCH],
                                                    int l32, int width) {
    if (MASKED) RowIO32M<T, NCH>::atomic_add(row, v, l32, width);
    else RowIO32<T, NCH>::atomic_add(row, v, l32);
  }
};

// Per-half (32-lane) sum: 4x row_shr + row_bcast15 leaves each half's total
// in its lane 31/63; one width-32 shuffle broadcasts it within the half.
__device__ __forceinline__ float half_sum_f32(float v) {
  typedef int i32;
  i32 x = __float_as_int(v);
#define DPPH_ADD(ctrl)                                                       \
  x = __float_as_int(__int_as_float(x) +                                     \
      __int_as_float(__builtin_amdgcn_update_dpp(0, x, ctrl, 0xF, 0xF, true)))
  DPPH_ADD(0x111);
  DPPH_ADD(0x112);
  DPPH_ADD(0x114);
  DPPH_ADD(0x118);
  DPPH_ADD(0x142);   // row_bcast:15 -> lane31 (half0) / lane63 (half1) total
#undef DPPH_ADD
  return __shfl(__int_as_float(x), 31, 32);
}

// ---------------------------------------------------------------------------
// sigmoid with the reference's +-MAX_EXP clip (mllib:281-302 semantics,
// exact sigmoid instead of the 1000-entry LUT approximation)
// ---------------------------------------------------------------------------
__device__ __forceinline__ float sigmoid_clipped(float f) {
  if (f > 6.0f) return 1.0f;
  if (f < -6.0f) return 0.0f;
  // precise expf (not __expf): one call per pair per wave — cost is
  // negligible and it keeps parity with the CPU oracle tight
  return 1.0f / (1.0f + expf(-f));
}

// reference getSigmoid parity mode (mllib:292-302): floor-indexed LUT
__device__ __forceinline__ float sigma_of(float f, const float* et, int etn) {
  if (et) {
    if (f > 6.0f) return 1.0f;
    if (f < -6.0f) return 0.0f;
    int i = (int)((f + 6.0f) * (etn / 12.0f));
    i = i >= etn ? etn - 1 : (i < 0 ? 0 : i);
    return et[i];
  }
  return sigmoid_clipped(f);
}

// Fast full-wave (64-lane) float sum: 4x row_shr + row_bcast15/31 DPP adds
// (bound_ctrl=0-fill), total in lane 63, broadcast via readlane.  Replaces
// the 6-step __shfl_xor chain (ds_bpermute + mask setup, ~7x more VALU).
__device__ __forceinline__ float wave_sum_f32(float v) {
  typedef int i32;
  i32 x = __float_as_int(v);
#define DPP_ADD(ctrl)                                                        \
  x = __float_as_int(__int_as_float(x) +                                     \
      __int_as_float(__builtin_amdgcn_update_dpp(0, x, ctrl, 0xF, 0xF, true)))
  DPP_ADD(0x111);   // row_shr:1
  DPP_ADD(0x112);   // row_shr:2
  DPP_ADD(0x114);   // row_shr:4
  DPP_ADD(0x118);   // row_shr:8
  DPP_ADD(0x142);   // row_bcast:15
  DPP_ADD(0x143);   // row_bcast:31
#undef DPP_ADD
  return __int_as_float(__builtin_amdgcn_readlane(x, 63));
}

struct KernelArgs {
  void* syn0;
  void* syn1;
  const int32_t* tokens;
  const int32_t* offsets;
  int64_t num_sentences;
  const uint32_t* keep_thr;   // nullptr = subsampling off
  const int32_t* table;
  uint32_t table_size;
  float alpha;
  int window;
  int n_neg;
  uint64_t seed;
  int64_t sent_id_base;
  int64_t stride;             // row stride in elements (= 64*NC)
  int ref_window;             // 0 canonical, 1 reference (B2) semantics
  int32_t atomic_below;       // rows < this use atomics (hot rows); rest plain
  // rows < atomic_floor take the plain path even in atomic modes: the
  // ultra-hot head (top ~100 Zipf rows) is hit by a double-digit share of
  // all negative draws, so atomics there serialize on a few cachelines —
  // the hybrid cliff.  Those rows are subsample-suppressed as
  // centers/contexts anyway; hogwild on them costs no measurable quality
  // (benchmarks/results.md round-2 sweep).
  int32_t atomic_floor;
  // HogBatch-style shared negatives (rng.py layout: k = NEG_BASE + i*n + s,
  // one draw set per position reused across its contexts) — opt-in,
  // config.shared_negatives.  Cuts per-position target-row traffic from
  // ~2b*(1+n) to ~2b+n rows (the contexts' negative rows stay LLC-hot).
  int shared_neg;
  const float* exp_table;     // non-null: reference LUT sigmoid parity mode
  int exp_table_size;
  int width;                  // valid elements per row (masked dim phases)
  // stats
  unsigned long long* d_pairs;
  unsigned long long* d_positives;
  unsigned long long* d_words;
  double* d_sum_fplus;
};

constexpr int kMaxSent = 1024;
constexpr int kWavesPerBlock = 4;
constexpr int kPosBlock = 96;   // kept positions per wave when grid is 2-D

// ---------------------------------------------------------------------------
// Shared sentence walker.  Subsample+compact into the wave's LDS buffer,
// then iterate positions/pairs in the normative order (rng.py), invoking
// Phase:  begin_position(c) -> bool want_pairs;
//         pair(tgt, label, pair_idx);
//         end_position(c).
// pair_idx counts emitted pairs within the sentence, identical on every
// rank/implementation — the spine of the dim-sharded engine.
// ---------------------------------------------------------------------------
// pos_lo/pos_hi bound the kept-position range this wave processes (position
// blocks let long sentences spread over many waves; RNG is counter-based so
// any wave can process any position).  pair_idx stays GLOBAL to the
// sentence: the walker enumerates all positions, skipping work outside the
// range but keeping the pair numbering identical (dim-sharded f indexing).
template <typename Phase>
__device__ __forceinline__ void walk_sentence_dev(
    const int32_t* __restrict__ tokens, int64_t off, int len, uint64_t base,
    const uint32_t* __restrict__ keep_thr, const int32_t* __restrict__ table,
    uint32_t table_size, int window, int n_neg, int ref_window,
    int shared_neg, int lane, int32_t* sent_lds, uint32_t* tgt_lds,
    Phase& ph, int pos_lo = 0, int pos_hi = 1 << 30) {
  // ---- subsample + wave compaction into LDS -----------------------------
  int L = 0;
  for (int p0 = 0; p0 < len; p0 += 64) {
    const int p = p0 + lane;
    bool keep = false;
    int32_t w = 0;
    if (p < len) {
      w = tokens[off + p];
      if (keep_thr) {
        const uint32_t u = draw_u32(base, (uint64_t)p);
        keep = u <= keep_thr[w];
      } else {
        keep = true;
      }
    }
    const uint64_t m = __ballot(keep);
    const int pos = __popcll(m & ((1ULL << lane) - 1ULL));
    if (keep) sent_lds[L + pos] = w;
    L += __popcll(m);
  }
  // (no __syncthreads needed: LDS buffers are private to this wave)

  int64_t pair_idx = 0;
  const int per_ctx = 1 + n_neg;
  for (int i = 0; i < L; ++i) {
    const int32_t c = sent_lds[i];
    const uint32_t u = draw_u32(base, kWinBase + (uint64_t)i);
    int lo, hi;
    if (!ref_window) {
      const int b = 1 + (int)(u % (uint32_t)window);
      lo = i - b < 0 ? 0 : i - b;
      hi = i + b >= L ? L - 1 : i + b;
    } else {
      const int b = (int)(u % (uint32_t)window);
      if (b == 0) { lo = i; hi = i; }
      else {
        lo = i - b < 0 ? 0 : i - b;
        hi = i + b - 1 >= L ? L - 1 : i + b - 1;
      }
    }
    if (!((lo < i) || (hi > i))) continue;
    const int n_ctx = (hi - lo + 1) - ((lo <= i && i <= hi) ? 1 : 0);
    // shared mode: slots = [n_ctx positives][n_neg shared negatives] —
    // ONE negative set per position (rng.py shared layout), vs the
    // per-context n_ctx*(1+n_neg) enumeration
    const int total_slots =
        shared_neg ? (n_ctx + n_neg) : n_ctx * per_ctx;
    if (i < pos_lo || i >= pos_hi) {
      // outside this wave's position block: advance the pair numbering
      // without touching rows.  Negative-collision discards still need the
      // draws; do them 64 slots at a time like the main path.
      for (int chunk = 0; chunk < total_slots; chunk += 64) {
        const int slot = chunk + lane;
        bool valid = false;
        if (slot < total_slots) {
          if (shared_neg) {
            if (slot < n_ctx) {
              valid = true;
            } else {
              const uint32_t un = draw_u32(
                  base, kNegBase + (uint64_t)i * (uint64_t)n_neg +
                            (uint64_t)(slot - n_ctx));
              valid = (table[un % table_size] != c);
            }
          } else {
            const int ctx_i = slot / per_ctx;
            const int s_in = slot - ctx_i * per_ctx;
            if (s_in == 0) {
              valid = true;
            } else {
              int j = lo + ctx_i;
              if (j >= i) ++j;
              const int32_t t = sent_lds[j];
              const uint64_t kbase = kNegBase +
                  (uint64_t)(i * (2 * window + 1) + (j - i + window)) *
                      (uint64_t)n_neg;
              const uint32_t un =
                  draw_u32(base, kbase + (uint64_t)(s_in - 1));
              valid = (table[un % table_size] != t);
            }
          }
        }
        pair_idx += __popcll(__ballot(valid));
      }
      continue;
    }
    ph.begin_position(c);
    // Materialize the target list 64 slots at a time (all lanes draw in
    // parallel — negative RNG + table gathers vectorize across the wave),
    // then the phase processes the compacted chunk serially with prefetch.
    for (int chunk = 0; chunk < total_slots; chunk += 64) {
      const int slot = chunk + lane;
      bool valid = false;
      uint32_t enc = 0;
      if (slot < total_slots) {
        if (shared_neg) {
          if (slot < n_ctx) {
            int j = lo + slot;
            if (j >= i) ++j;                   // skip the center position
            enc = (uint32_t)sent_lds[j] | 0x80000000u;
            valid = true;
          } else {
            const uint32_t un = draw_u32(
                base, kNegBase + (uint64_t)i * (uint64_t)n_neg +
                          (uint64_t)(slot - n_ctx));
            const int32_t neg = table[un % table_size];
            valid = (neg != c);                // discard center collision
            enc = (uint32_t)neg;
          }
        } else {
          const int ctx_i = slot / per_ctx;
          const int s_in = slot - ctx_i * per_ctx;
          int j = lo + ctx_i;
          if (j >= i) ++j;                     // skip the center position
          const int32_t t = sent_lds[j];
          if (s_in == 0) {
            enc = (uint32_t)t | 0x80000000u;   // positive: bit 31 set
            valid = true;
          } else {
            const uint64_t kbase = kNegBase +
                (uint64_t)(i * (2 * window + 1) + (j - i + window)) *
                    (uint64_t)n_neg;
            const uint32_t un = draw_u32(base, kbase + (uint64_t)(s_in - 1));
            const int32_t neg = table[un % table_size];
            valid = (neg != t);                // discard colliding negative
            enc = (uint32_t)neg;
          }
        }
      }
      const uint64_t m = __ballot(valid);
      const int pos = __popcll(m & ((1ULL << lane) - 1ULL));
      if (valid) tgt_lds[pos] = enc;
      const int count = __popcll(m);
      if (count > 0) {
        ph.process_pairs(tgt_lds, count, pair_idx);
        pair_idx += count;
      }
    }
    ph.end_position(c);
  }
}

// ---- Phase: fully fused train (dot + sigmoid + update in one pass) -------
// process_pairs: 2-deep software pipeline — the next target row's loads are
// issued before the current pair's dot/update consumes its row, hiding the
// HBM/LLC gather latency under compute (guide G7/G15; explicit ping-pong
// buffers keep indices compile-time, rule 20).
template <typename T, int NC, bool ATOMIC>
struct TrainPhase {
  T* syn0;
  T* syn1;
  int64_t stride;
  float alpha;
  int lane;
  int32_t atomic_below;
  int32_t atomic_floor;
  const float* exp_table;
  int exp_table_size;
  // per-position state
  T* c_ptr;
  int32_t c_idx;
  float c_row[NC];
  float grad[NC];
  // stats
  unsigned long long w_pairs = 0, w_pos = 0, w_words = 0;
  float w_fplus = 0.0f;

  __device__ __forceinline__ void begin_position(int32_t c) {
    c_idx = c;
    c_ptr = syn0 + (int64_t)c * stride;
    RowIO<T, NC>::load(c_ptr, c_row, lane);
#pragma unroll
    for (int k = 0; k < NC; ++k) grad[k] = 0.0f;
  }

  __device__ __forceinline__ T* rowptr(uint32_t enc) const {
    return syn1 + (int64_t)(enc & 0x7FFFFFFFu) * stride;
  }

  __device__ __forceinline__ void do_pair(uint32_t enc, float (&t_row)[NC],
                                          T* t_ptr) {
    const float label = (enc & 0x80000000u) ? 1.0f : 0.0f;
    float f = 0.0f;
#pragma unroll
    for (int k = 0; k < NC; ++k) f += c_row[k] * t_row[k];
    f = wave_sum_f32(f);
    const float g = (label - sigma_of(f, exp_table, exp_table_size)) * alpha;
    // atomic_below < 0: atomics for POSITIVE pairs only (they drive the
    // learning signal; negatives self-limit) — ~1/6 of the update traffic
    const int32_t rid = (int32_t)(enc & 0x7FFFFFFFu);
    const bool use_atomic =
        ATOMIC && (atomic_below < 0 ? (label > 0.5f)
                                    : (rid < atomic_below && rid >= atomic_floor));
    if (use_atomic) {
      // t_row doubles as the delta buffer (it is dead after this pair) —
      // no extra NC registers for the atomic path
#pragma unroll
      for (int k = 0; k < NC; ++k) {
        grad[k] += g * t_row[k];
        t_row[k] = g * c_row[k];
      }
      RowIO<T, NC>::atomic_add(t_ptr, t_row, lane);
    } else {
#pragma unroll
      for (int k = 0; k < NC; ++k) {
        grad[k] += g * t_row[k];
        t_row[k] += g * c_row[k];
      }
      RowIO<T, NC>::store(t_ptr, t_row, lane);
    }
    ++w_pairs;
    if (enc & 0x80000000u) {
      ++w_pos;
      w_fplus += f;
    }
  }

  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t) {
    if (NC >= 16) {
      // wide rows (dim >= ~1024): the 2-deep prefetch costs an extra NC
      // registers (b1) that collapse occupancy below what HBM latency
      // hiding needs — a plain load-use loop keeps ~2 more waves/SIMD in
      // flight and wins (A/B in benchmarks/results.md round 2)
      for (int k = 0; k < count; ++k) {
        const uint32_t e = tl[k];
        T* p = rowptr(e);
        float b[NC];
        RowIO<T, NC>::load(p, b, lane);
        do_pair(e, b, p);
      }
      return;
    }
    float b0[NC], b1[NC];
    uint32_t e0 = tl[0], e1 = 0;
    T* p0 = rowptr(e0);
    T* p1 = nullptr;
    RowIO<T, NC>::load(p0, b0, lane);
    int k = 0;
    for (;;) {
      if (k + 1 < count) {
        e1 = tl[k + 1];
        p1 = rowptr(e1);
        RowIO<T, NC>::load(p1, b1, lane);
      }
      do_pair(e0, b0, p0);
      if (++k >= count) break;
      // adjacent pairs on the same row: the prefetch predates this pair's
      // update — refresh so sequential semantics (and oracle parity) hold
      if (((e1 ^ e0) & 0x7FFFFFFFu) == 0) RowIO<T, NC>::load(p1, b1, lane);
      if (k + 1 < count) {
        e0 = tl[k + 1];
        p0 = rowptr(e0);
        RowIO<T, NC>::load(p0, b0, lane);
      }
      do_pair(e1, b1, p1);
      if (++k >= count) break;
      if (((e0 ^ e1) & 0x7FFFFFFFu) == 0) RowIO<T, NC>::load(p0, b0, lane);
    }
  }

  __device__ __forceinline__ void end_position(int32_t) {
    // center row update (hogwild: re-read current value, add, store)
    if (ATOMIC && (atomic_below < 0 ||
                   (c_idx < atomic_below && c_idx >= atomic_floor))) {
      RowIO<T, NC>::atomic_add(c_ptr, grad, lane);
    } else {
      float cur[NC];
      RowIO<T, NC>::load(c_ptr, cur, lane);
#pragma unroll
      for (int k = 0; k < NC; ++k) cur[k] += grad[k];
      RowIO<T, NC>::store(c_ptr, cur, lane);
    }
    ++w_words;
  }
};

template <typename T, int NC, bool ATOMIC>
__global__ __launch_bounds__(64 * kWavesPerBlock) void sgns_train_kernel(KernelArgs a) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;   // 1 in serial mode, else 4
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;

  TrainPhase<T, NC, ATOMIC> ph{};
  ph.syn0 = (T*)a.syn0;
  ph.syn1 = (T*)a.syn1;
  ph.stride = a.stride;
  ph.alpha = a.alpha;
  ph.lane = lane;
  ph.atomic_below = a.atomic_below;
  ph.atomic_floor = a.atomic_floor;
  ph.exp_table = a.exp_table;
  ph.exp_table_size = a.exp_table_size;

  // gridDim.y > 1 splits each sentence into position blocks of kPosBlock
  // kept positions so long sentences fill the chip (hogwild across blocks,
  // same class as across sentences)
  const int pos_lo = (int)blockIdx.y * kPosBlock;
  const int pos_hi = gridDim.y > 1 ? pos_lo + kPosBlock : (1 << 30);
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph, pos_lo,
                      pos_hi);
  }

  // ---- stats (device-scope atomics, once per wave) -----------------------
  if (lane == 0 && a.d_pairs) {
    atomicAdd(a.d_pairs, ph.w_pairs);
    atomicAdd(a.d_positives, ph.w_pos);
    atomicAdd(a.d_words, ph.w_words);
    atomicAdd(a.d_sum_fplus, (double)ph.w_fplus);
  }
}

// ---- Phase: fused train, TWO pairs per wave (32 lanes each) --------------
// Each half-wave owns one pair's target row: 2x rows in flight per wave at
// the same instruction count (halves process their own data in the same
// VALU instructions).  Pair 2k+1 reads rows before pair 2k's update lands —
// one extra pair of hogwild staleness; the serial/parity path keeps the
// 64-lane kernel.  Center grads accumulate per half and are combined with
// one cross-half shuffle at position end.
// PIPE: software-pipeline the half-wave pair blocks 2-deep — the next
// block's row loads issue BEFORE this block's update, so the update's
// atomic/store traffic (which clogs vmcnt) no longer delays the next
// load's waitcnt.  Costs NCH extra VGPRs; one extra block of hogwild
// staleness (same race class as the half-split itself).
template <typename T, int NCH, bool ATOMIC, bool PIPE = false>
struct TrainPhase2 {
  T* syn0;
  T* syn1;
  int64_t stride;
  float alpha;
  int lane;     // 0..63
  int l32;      // lane & 31
  int half;     // lane >> 5
  int32_t atomic_below;
  int32_t atomic_floor;
  const float* exp_table;
  int exp_table_size;
  T* c_ptr;
  int32_t c_idx;
  float c_row[NCH];
  float grad[NCH];
  uint32_t w_pairs = 0, w_pos = 0, w_words = 0;
  float w_fplus = 0.0f;

  __device__ __forceinline__ void begin_position(int32_t c) {
    c_idx = c;
    c_ptr = syn0 + (int64_t)c * stride;
    RowIO32<T, NCH>::load(c_ptr, c_row, l32);
#pragma unroll
    for (int k = 0; k < NCH; ++k) grad[k] = 0.0f;
  }

  __device__ __forceinline__ void do_block(uint32_t enc, bool active,
                                           float (&t_row)[NCH], T* t_ptr) {
    float f = 0.0f;
#pragma unroll
    for (int m = 0; m < NCH; ++m) f += c_row[m] * t_row[m];
    f = half_sum_f32(f);
    const float label = (enc & 0x80000000u) ? 1.0f : 0.0f;
    const float g0 = (label - sigma_of(f, exp_table, exp_table_size)) * alpha;
    const float g = active ? g0 : 0.0f;     // idle half: zero contribution
#pragma unroll
    for (int m = 0; m < NCH; ++m) grad[m] += g * t_row[m];
    if (active) {
      const int32_t rid = (int32_t)(enc & 0x7FFFFFFFu);
      const bool use_atomic =
          ATOMIC && (atomic_below < 0 ? (label > 0.5f)
                                      : (rid < atomic_below && rid >= atomic_floor));
      if (use_atomic) {
#pragma unroll
        for (int m = 0; m < NCH; ++m) t_row[m] = g * c_row[m];
        RowIO32<T, NCH>::atomic_add(t_ptr, t_row, l32);
      } else {
#pragma unroll
        for (int m = 0; m < NCH; ++m) t_row[m] += g * c_row[m];
        RowIO32<T, NCH>::store(t_ptr, t_row, l32);
      }
      ++w_pairs;
      if (label > 0.5f) {
        ++w_pos;
        w_fplus += f;
      }
    }
  }

  __device__ __forceinline__ void load_block(const uint32_t* tl, int count,
                                             int k, uint32_t& enc, T*& ptr,
                                             float (&buf)[NCH], bool& act) {
    const int my = k + half;
    act = my < count;
    enc = tl[act ? my : k];
    ptr = syn1 + (int64_t)(enc & 0x7FFFFFFFu) * stride;
    RowIO32<T, NCH>::load(ptr, buf, l32);
  }

  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t) {
    if (!PIPE) {
      for (int k = 0; k < count; k += 2) {
        uint32_t enc;
        T* t_ptr;
        float t_row[NCH];
        bool active;
        load_block(tl, count, k, enc, t_ptr, t_row, active);
        do_block(enc, active, t_row, t_ptr);
      }
      return;
    }
    // 2-deep ping-pong (TrainPhase's pattern, per half-wave block)
    uint32_t e0, e1;
    T *p0, *p1;
    float b0[NCH], b1[NCH];
    bool a0, a1;
    load_block(tl, count, 0, e0, p0, b0, a0);
    for (int k = 0;; k += 4) {
      if (k + 2 < count) load_block(tl, count, k + 2, e1, p1, b1, a1);
      do_block(e0, a0, b0, p0);
      if (k + 2 >= count) break;
      if (k + 4 < count) load_block(tl, count, k + 4, e0, p0, b0, a0);
      do_block(e1, a1, b1, p1);
      if (k + 4 >= count) break;
    }
  }

  __device__ __forceinline__ void end_position(int32_t) {
    // combine the halves' grads (same elements live at lane l and l+32)
#pragma unroll
    for (int m = 0; m < NCH; ++m) grad[m] += __shfl_xor(grad[m], 32, 64);
    if (ATOMIC && (atomic_below < 0 ||
                   (c_idx < atomic_below && c_idx >= atomic_floor))) {
      if (half == 0) RowIO32<T, NCH>::atomic_add(c_ptr, grad, l32);
    } else if (half == 0) {
      // one half does the RMW — the other would duplicate identical bytes
      float cur[NCH];
      RowIO32<T, NCH>::load(c_ptr, cur, l32);
#pragma unroll
      for (int m = 0; m < NCH; ++m) cur[m] += grad[m];
      RowIO32<T, NCH>::store(c_ptr, cur, l32);
    }
    ++w_words;
  }
};

template <typename T, int NCH, bool ATOMIC, bool PIPE = false>
__global__ __launch_bounds__(64 * kWavesPerBlock) void sgns_train2_kernel(KernelArgs a) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;

  TrainPhase2<T, NCH, ATOMIC, PIPE> ph{};
  ph.syn0 = (T*)a.syn0;
  ph.syn1 = (T*)a.syn1;
  ph.stride = a.stride;
  ph.alpha = a.alpha;
  ph.lane = lane;
  ph.l32 = lane & 31;
  ph.half = lane >> 5;
  ph.atomic_below = a.atomic_below;
  ph.atomic_floor = a.atomic_floor;
  ph.exp_table = a.exp_table;
  ph.exp_table_size = a.exp_table_size;

  const int pos_lo = (int)blockIdx.y * kPosBlock;
  const int pos_hi = gridDim.y > 1 ? pos_lo + kPosBlock : (1 << 30);
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph, pos_lo,
                      pos_hi);
  }
  // combine half stats then one atomic from lane 0
  unsigned int p2 = ph.w_pairs + __shfl_xor(ph.w_pairs, 32, 64);
  unsigned int o2 = ph.w_pos + __shfl_xor(ph.w_pos, 32, 64);
  float f2 = ph.w_fplus + __shfl_xor(ph.w_fplus, 32, 64);
  if (lane == 0 && a.d_pairs) {
    atomicAdd(a.d_pairs, (unsigned long long)p2);
    atomicAdd(a.d_positives, (unsigned long long)o2);
    atomicAdd(a.d_words, (unsigned long long)ph.w_words);
    atomicAdd(a.d_sum_fplus, (double)f2);
  }
}

// ---- Phase: fused train, FOUR pairs per wave (16-lane quarters) ----------
template <typename T, int NCQ, bool ATOMIC>
struct TrainPhase4 {
  T* syn0;
  T* syn1;
  int64_t stride;
  float alpha;
  int l16;      // lane & 15
  int quarter;  // lane >> 4
  int32_t atomic_below;
  int32_t atomic_floor;
  const float* exp_table;
  int exp_table_size;
  T* c_ptr;
  int32_t c_idx;
  float c_row[NCQ];
  float grad[NCQ];
  uint32_t w_pairs = 0, w_pos = 0, w_words = 0;
  float w_fplus = 0.0f;

  __device__ __forceinline__ void begin_position(int32_t c) {
    c_idx = c;
    c_ptr = syn0 + (int64_t)c * stride;
    RowIO16<T, NCQ>::load(c_ptr, c_row, l16);
#pragma unroll
    for (int k = 0; k < NCQ; ++k) grad[k] = 0.0f;
  }

  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t) {
    for (int k = 0; k < count; k += 4) {
      const int my = k + quarter;
      const bool active = my < count;
      const uint32_t enc = tl[active ? my : k];
      T* t_ptr = syn1 + (int64_t)(enc & 0x7FFFFFFFu) * stride;
      float t_row[NCQ];
      RowIO16<T, NCQ>::load(t_ptr, t_row, l16);
      float f = 0.0f;
#pragma unroll
      for (int m = 0; m < NCQ; ++m) f += c_row[m] * t_row[m];
      f = quarter_sum_f32(f);
      const float label = (enc & 0x80000000u) ? 1.0f : 0.0f;
      const float g0 = (label - sigma_of(f, exp_table, exp_table_size)) * alpha;
      const float g = active ? g0 : 0.0f;
#pragma unroll
      for (int m = 0; m < NCQ; ++m) grad[m] += g * t_row[m];
      if (active) {
        const int32_t rid = (int32_t)(enc & 0x7FFFFFFFu);
        const bool use_atomic =
            ATOMIC && rid < atomic_below && rid >= atomic_floor;
        if (use_atomic) {
#pragma unroll
          for (int m = 0; m < NCQ; ++m) t_row[m] = g * c_row[m];
          RowIO16<T, NCQ>::atomic_add(t_ptr, t_row, l16);
        } else {
#pragma unroll
          for (int m = 0; m < NCQ; ++m) t_row[m] += g * c_row[m];
          RowIO16<T, NCQ>::store(t_ptr, t_row, l16);
        }
        ++w_pairs;
        if (label > 0.5f) {
          ++w_pos;
          w_fplus += f;
        }
      }
    }
  }

  __device__ __forceinline__ void end_position(int32_t) {
#pragma unroll
    for (int m = 0; m < NCQ; ++m) {
      grad[m] += __shfl_xor(grad[m], 16, 64);
      grad[m] += __shfl_xor(grad[m], 32, 64);
    }
    if (ATOMIC && c_idx < atomic_below && c_idx >= atomic_floor) {
      if (quarter == 0) RowIO16<T, NCQ>::atomic_add(c_ptr, grad, l16);
    } else if (quarter == 0) {
      float cur[NCQ];
      RowIO16<T, NCQ>::load(c_ptr, cur, l16);
#pragma unroll
      for (int m = 0; m < NCQ; ++m) cur[m] += grad[m];
      RowIO16<T, NCQ>::store(c_ptr, cur, l16);
    }
    ++w_words;
  }
};

template <typename T, int NCQ, bool ATOMIC>
__global__ __launch_bounds__(64 * kWavesPerBlock) void sgns_train4_kernel(KernelArgs a) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;

  TrainPhase4<T, NCQ, ATOMIC> ph{};
  ph.syn0 = (T*)a.syn0;
  ph.syn1 = (T*)a.syn1;
  ph.stride = a.stride;
  ph.alpha = a.alpha;
  ph.l16 = lane & 15;
  ph.quarter = lane >> 4;
  ph.atomic_below = a.atomic_below;
  ph.atomic_floor = a.atomic_floor;
  ph.exp_table = a.exp_table;
  ph.exp_table_size = a.exp_table_size;

  const int pos_lo = (int)blockIdx.y * kPosBlock;
  const int pos_hi = gridDim.y > 1 ? pos_lo + kPosBlock : (1 << 30);
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph, pos_lo,
                      pos_hi);
  }
  unsigned int p2 = ph.w_pairs, o2 = ph.w_pos;
  float f2 = ph.w_fplus;
  p2 += __shfl_xor(p2, 16, 64); p2 += __shfl_xor(p2, 32, 64);
  o2 += __shfl_xor(o2, 16, 64); o2 += __shfl_xor(o2, 32, 64);
  f2 += __shfl_xor(f2, 16, 64); f2 += __shfl_xor(f2, 32, 64);
  if (lane == 0 && a.d_pairs) {
    atomicAdd(a.d_pairs, (unsigned long long)p2);
    atomicAdd(a.d_positives, (unsigned long long)o2);
    atomicAdd(a.d_words, (unsigned long long)ph.w_words);
    atomicAdd(a.d_sum_fplus, (double)f2);
  }
}

// ---- Phase: count pairs (dim-sharded phase 0) ----------------------------
struct CountPhase {
  int64_t pairs = 0;
  __device__ __forceinline__ void begin_position(int32_t) {}
  __device__ __forceinline__ void process_pairs(const uint32_t*, int count,
                                                int64_t) {
    pairs += count;
  }
  __device__ __forceinline__ void end_position(int32_t) {}
};

// ---- plan emission (row-sharded engine): materialise the walker's pair
// enumeration — counter-based RNG, identical to the fused kernel's — into
// explicit plan arrays.  Each sentence is written by exactly one wave at
// offsets fixed by pair_offsets, so the output is deterministic at any
// grid size.  out_start marks the first pair of each center position
// (group boundaries). -------------------------------------------------------
struct PlanEmitPhase {
  int32_t* out_target;
  float* out_label;
  uint8_t* out_start;
  int32_t* out_center;
  int64_t base;          // pair_offsets[s]
  int lane;
  int32_t c_word;
  bool at_pos_start;
  __device__ __forceinline__ void begin_position(int32_t c) {
    c_word = c;
    at_pos_start = true;
  }
  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t idx_base) {
    if (lane < count) {
      const uint32_t enc = tl[lane];
      const int64_t idx = base + idx_base + lane;
      out_target[idx] = (int32_t)(enc & 0x7FFFFFFFu);
      out_label[idx] = (enc & 0x80000000u) ? 1.0f : 0.0f;
      out_center[idx] = c_word;
      out_start[idx] = (at_pos_start && lane == 0) ? 1 : 0;
    }
    at_pos_start = false;
  }
  __device__ __forceinline__ void end_position(int32_t) {}
};

__global__ __launch_bounds__(64 * kWavesPerBlock) void plan_emit_kernel(
    KernelArgs a, const int64_t* __restrict__ pair_offsets,
    int32_t* __restrict__ out_target, float* __restrict__ out_label,
    uint8_t* __restrict__ out_start, int32_t* __restrict__ out_center) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    PlanEmitPhase ph{};
    ph.out_target = out_target;
    ph.out_label = out_label;
    ph.out_start = out_start;
    ph.out_center = out_center;
    ph.base = pair_offsets[s];
    ph.lane = lane;
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph);
  }
}

__global__ __launch_bounds__(64 * kWavesPerBlock) void count_pairs_kernel(
    KernelArgs a, int64_t* __restrict__ counts) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    CountPhase ph{};
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph);
    if (lane == 0) counts[s] = ph.pairs;
  }
}

// ---- Phase: partial dots over a dim-slice (dim-sharded phase 1) ----------
template <typename T, int NC>
struct DotPhase {
  const T* syn0;
  const T* syn1;
  int64_t stride;
  float* f_base;   // f output, sentence-local
  int lane;
  float c_row[NC];
  __device__ __forceinline__ void begin_position(int32_t c) {
    RowIO<T, NC>::load(syn0 + (int64_t)c * stride, c_row, lane);
  }
  __device__ __forceinline__ const T* rowptr(uint32_t enc) const {
    return syn1 + (int64_t)(enc & 0x7FFFFFFFu) * stride;
  }
  __device__ __forceinline__ void dot_one(const float (&t_row)[NC], int64_t idx) {
    float f = 0.0f;
#pragma unroll
    for (int k = 0; k < NC; ++k) f += c_row[k] * t_row[k];
    f = wave_sum_f32(f);
    if (lane == 0) f_base[idx] = f;
  }
  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t idx_base) {
    float b0[NC], b1[NC];
    const T* p0 = rowptr(tl[0]);
    RowIO<T, NC>::load(p0, b0, lane);
    int k = 0;
    for (;;) {
      if (k + 1 < count) RowIO<T, NC>::load(rowptr(tl[k + 1]), b1, lane);
      dot_one(b0, idx_base + k);
      if (++k >= count) break;
      if (k + 1 < count) RowIO<T, NC>::load(rowptr(tl[k + 1]), b0, lane);
      dot_one(b1, idx_base + k);
      if (++k >= count) break;
    }
  }
  __device__ __forceinline__ void end_position(int32_t) {}
};

template <typename T, int NC>
__global__ __launch_bounds__(64 * kWavesPerBlock) void dots_slice_kernel(
    KernelArgs a, const int64_t* __restrict__ pair_offsets,
    float* __restrict__ f_out) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    DotPhase<T, NC> ph{};
    ph.syn0 = (const T*)a.syn0;
    ph.syn1 = (const T*)a.syn1;
    ph.stride = a.stride;
    ph.f_base = f_out + pair_offsets[s];
    ph.lane = lane;
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph);
  }
}

// ---- Phase: slice update from allreduced dots (dim-sharded phase 2) ------
// With f_loc != null, the stale allreduced dot is freshened by extrapolating
// this rank's local drift:
//   f_used = f_total + world * (local_partial_now - local_partial_at_pass1)
template <typename T, int NC, bool ATOMIC>
struct UpdateSlicePhase {
  T* syn0;
  T* syn1;
  int64_t stride;
  const float* f_base;   // full (summed) dots, sentence-local
  const float* f_loc;    // pass-1 local partials (null = correction off)
  float world_scale;
  float alpha;
  int lane;
  int32_t atomic_below;
  int32_t atomic_floor;
  const float* exp_table;
  int exp_table_size;
  T* c_ptr;
  int32_t c_idx;
  float c_row[NC];
  float grad[NC];
  unsigned long long w_pairs = 0, w_pos = 0, w_words = 0;
  float w_fplus = 0.0f;
  __device__ __forceinline__ void begin_position(int32_t c) {
    c_idx = c;
    c_ptr = syn0 + (int64_t)c * stride;
    RowIO<T, NC>::load(c_ptr, c_row, lane);
#pragma unroll
    for (int k = 0; k < NC; ++k) grad[k] = 0.0f;
  }
  __device__ __forceinline__ T* rowptr(uint32_t enc) const {
    return syn1 + (int64_t)(enc & 0x7FFFFFFFu) * stride;
  }
  __device__ __forceinline__ void do_pair(uint32_t enc, float (&t_row)[NC],
                                          T* t_ptr, int64_t idx) {
    const float label = (enc & 0x80000000u) ? 1.0f : 0.0f;
    float f = f_base[idx];
    if (f_loc) {
      float fresh = 0.0f;
#pragma unroll
      for (int k = 0; k < NC; ++k) fresh += c_row[k] * t_row[k];
      fresh = wave_sum_f32(fresh);
      f += world_scale * (fresh - f_loc[idx]);
    }
    const float g = (label - sigma_of(f, exp_table, exp_table_size)) * alpha;
    // atomic_below < 0: atomics for POSITIVE pairs only (they drive the
    // learning signal; negatives self-limit) — ~1/6 of the update traffic
    const int32_t rid = (int32_t)(enc & 0x7FFFFFFFu);
    const bool use_atomic =
        ATOMIC && (atomic_below < 0 ? (label > 0.5f)
                                    : (rid < atomic_below && rid >= atomic_floor));
    if (use_atomic) {
      // t_row doubles as the delta buffer (it is dead after this pair) —
      // no extra NC registers for the atomic path
#pragma unroll
      for (int k = 0; k < NC; ++k) {
        grad[k] += g * t_row[k];
        t_row[k] = g * c_row[k];
      }
      RowIO<T, NC>::atomic_add(t_ptr, t_row, lane);
    } else {
#pragma unroll
      for (int k = 0; k < NC; ++k) {
        grad[k] += g * t_row[k];
        t_row[k] += g * c_row[k];
      }
      RowIO<T, NC>::store(t_ptr, t_row, lane);
    }
    ++w_pairs;
    if (enc & 0x80000000u) {
      ++w_pos;
      w_fplus += f;
    }
  }
  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t idx_base) {
    float b0[NC], b1[NC];
    uint32_t e0 = tl[0], e1 = 0;
    T* p0 = rowptr(e0);
    T* p1 = nullptr;
    RowIO<T, NC>::load(p0, b0, lane);
    int k = 0;
    for (;;) {
      if (k + 1 < count) {
        e1 = tl[k + 1];
        p1 = rowptr(e1);
        RowIO<T, NC>::load(p1, b1, lane);
      }
      do_pair(e0, b0, p0, idx_base + k);
      if (++k >= count) break;
      if (((e1 ^ e0) & 0x7FFFFFFFu) == 0) RowIO<T, NC>::load(p1, b1, lane);
      if (k + 1 < count) {
        e0 = tl[k + 1];
        p0 = rowptr(e0);
        RowIO<T, NC>::load(p0, b0, lane);
      }
      do_pair(e1, b1, p1, idx_base + k);
      if (++k >= count) break;
      if (((e0 ^ e1) & 0x7FFFFFFFu) == 0) RowIO<T, NC>::load(p0, b0, lane);
    }
  }
  __device__ __forceinline__ void end_position(int32_t) {
    if (ATOMIC && c_idx < atomic_below && c_idx >= atomic_floor) {
      RowIO<T, NC>::atomic_add(c_ptr, grad, lane);
    } else {
      float cur[NC];
      RowIO<T, NC>::load(c_ptr, cur, lane);
#pragma unroll
      for (int k = 0; k < NC; ++k) cur[k] += grad[k];
      RowIO<T, NC>::store(c_ptr, cur, lane);
    }
    ++w_words;
  }
};

template <typename T, int NC, bool ATOMIC>
__global__ __launch_bounds__(64 * kWavesPerBlock) void update_slice_kernel(
    KernelArgs a, const int64_t* __restrict__ pair_offsets,
    const float* __restrict__ f_in, const float* __restrict__ f_loc,
    float world_scale) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  UpdateSlicePhase<T, NC, ATOMIC> ph{};
  ph.syn0 = (T*)a.syn0;
  ph.syn1 = (T*)a.syn1;
  ph.stride = a.stride;
  ph.alpha = a.alpha;
  ph.lane = lane;
  ph.world_scale = world_scale;
  ph.atomic_below = a.atomic_below;
  ph.atomic_floor = a.atomic_floor;
  ph.exp_table = a.exp_table;
  ph.exp_table_size = a.exp_table_size;
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    ph.f_base = f_in + pair_offsets[s];
    ph.f_loc = f_loc ? f_loc + pair_offsets[s] : nullptr;
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph);
  }
  if (lane == 0 && a.d_pairs) {
    atomicAdd(a.d_pairs, ph.w_pairs);
    atomicAdd(a.d_positives, ph.w_pos);
    atomicAdd(a.d_words, ph.w_words);
    atomicAdd(a.d_sum_fplus, (double)ph.w_fplus);
  }
}

// ---- Phase: dim-sharded dots / update, TWO pairs per wave ---------------
template <typename T, int NCH, bool MASKED>
struct DotPhase2 {
  const T* syn0;
  const T* syn1;
  int64_t stride;
  float* f_base;
  int l32;
  int half;
  int width;
  float c_row[NCH];
  __device__ __forceinline__ void begin_position(int32_t c) {
    RIO32<T, NCH, MASKED>::load(syn0 + (int64_t)c * stride, c_row, l32,
                                width);
  }
  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t idx_base) {
    for (int k = 0; k < count; k += 2) {
      const int my = k + half;
      const bool active = my < count;
      const uint32_t enc = tl[active ? my : k];
      const T* t_ptr = syn1 + (int64_t)(enc & 0x7FFFFFFFu) * stride;
      float t_row[NCH];
      RIO32<T, NCH, MASKED>::load(t_ptr, t_row, l32, width);
      float f = 0.0f;
#pragma unroll
      for (int m = 0; m < NCH; ++m) f += c_row[m] * t_row[m];
      f = half_sum_f32(f);
      if (active && l32 == 0) f_base[idx_base + my] = f;
    }
  }
  __device__ __forceinline__ void end_position(int32_t) {}
};

template <typename T, int NCH, bool MASKED>
__global__ __launch_bounds__(64 * kWavesPerBlock) void dots_slice2_kernel(
    KernelArgs a, const int64_t* __restrict__ pair_offsets,
    float* __restrict__ f_out) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    DotPhase2<T, NCH, MASKED> ph{};
    ph.syn0 = (const T*)a.syn0;
    ph.syn1 = (const T*)a.syn1;
    ph.stride = a.stride;
    ph.f_base = f_out + pair_offsets[s];
    ph.l32 = lane & 31;
    ph.half = lane >> 5;
    ph.width = a.width;
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph);
  }
}

template <typename T, int NCH, bool ATOMIC, bool MASKED, bool PIPE = false>
struct UpdateSlicePhase2 {
  T* syn0;
  T* syn1;
  int64_t stride;
  const float* f_base;
  const float* f_loc;
  float world_scale;
  float alpha;
  int l32;
  int half;
  int width;
  int32_t atomic_below;
  int32_t atomic_floor;
  const float* exp_table;
  int exp_table_size;
  T* c_ptr;
  int32_t c_idx;
  float c_row[NCH];
  float grad[NCH];
  uint32_t w_pairs = 0, w_pos = 0, w_words = 0;
  float w_fplus = 0.0f;
  __device__ __forceinline__ void begin_position(int32_t c) {
    c_idx = c;
    c_ptr = syn0 + (int64_t)c * stride;
    RIO32<T, NCH, MASKED>::load(c_ptr, c_row, l32, width);
#pragma unroll
    for (int m = 0; m < NCH; ++m) grad[m] = 0.0f;
  }
  __device__ __forceinline__ void load_blk(const uint32_t* tl, int count,
                                           int k, uint32_t& enc, T*& ptr,
                                           float (&buf)[NCH], bool& act) {
    const int my = k + half;
    act = my < count;
    enc = tl[act ? my : k];
    ptr = syn1 + (int64_t)(enc & 0x7FFFFFFFu) * stride;
    RIO32<T, NCH, MASKED>::load(ptr, buf, l32, width);
  }

  __device__ __forceinline__ void do_blk(uint32_t enc, T* t_ptr,
                                         float (&t_row)[NCH], bool active,
                                         int64_t idx_base, int k) {
    const int my = k + half;
    float f = f_base[idx_base + (active ? my : k)];
    if (f_loc) {
      float fresh = 0.0f;
#pragma unroll
      for (int m = 0; m < NCH; ++m) fresh += c_row[m] * t_row[m];
      fresh = half_sum_f32(fresh);
      f += world_scale * (fresh - f_loc[idx_base + (active ? my : k)]);
    }
    const float label = (enc & 0x80000000u) ? 1.0f : 0.0f;
    const float g0 = (label - sigma_of(f, exp_table, exp_table_size)) * alpha;
    const float g = active ? g0 : 0.0f;
#pragma unroll
    for (int m = 0; m < NCH; ++m) grad[m] += g * t_row[m];
    if (active) {
      const int32_t rid = (int32_t)(enc & 0x7FFFFFFFu);
      const bool use_atomic =
          ATOMIC && rid < atomic_below && rid >= atomic_floor;
      if (use_atomic) {
#pragma unroll
        for (int m = 0; m < NCH; ++m) t_row[m] = g * c_row[m];
        RIO32<T, NCH, MASKED>::atomic_add(t_ptr, t_row, l32, width);
      } else {
#pragma unroll
        for (int m = 0; m < NCH; ++m) t_row[m] += g * c_row[m];
        RIO32<T, NCH, MASKED>::store(t_ptr, t_row, l32, width);
      }
      ++w_pairs;
      if (label > 0.5f) {
        ++w_pos;
        w_fplus += f;
      }
    }
  }

  __device__ __forceinline__ void process_pairs(const uint32_t* tl, int count,
                                                int64_t idx_base) {
    if (!PIPE) {
      for (int k = 0; k < count; k += 2) {
        uint32_t enc;
        T* t_ptr;
        float t_row[NCH];
        bool active;
        load_blk(tl, count, k, enc, t_ptr, t_row, active);
        do_blk(enc, t_ptr, t_row, active, idx_base, k);
      }
      return;
    }
    if (count <= 0) return;
    uint32_t e0, e1;
    T *p0, *p1;
    float b0[NCH], b1[NCH];
    bool a0, a1;
    load_blk(tl, count, 0, e0, p0, b0, a0);
    for (int k = 0;; k += 4) {
      if (k + 2 < count) load_blk(tl, count, k + 2, e1, p1, b1, a1);
      do_blk(e0, p0, b0, a0, idx_base, k);
      if (k + 2 >= count) break;
      if (k + 4 < count) load_blk(tl, count, k + 4, e0, p0, b0, a0);
      do_blk(e1, p1, b1, a1, idx_base, k + 2);
      if (k + 4 >= count) break;
    }
  }
  __device__ __forceinline__ void end_position(int32_t) {
#pragma unroll
    for (int m = 0; m < NCH; ++m) grad[m] += __shfl_xor(grad[m], 32, 64);
    if (ATOMIC && c_idx < atomic_below && c_idx >= atomic_floor) {
      if (half == 0)
        RIO32<T, NCH, MASKED>::atomic_add(c_ptr, grad, l32, width);
    } else if (half == 0) {
      float cur[NCH];
      RIO32<T, NCH, MASKED>::load(c_ptr, cur, l32, width);
#pragma unroll
      for (int m = 0; m < NCH; ++m) cur[m] += grad[m];
      RIO32<T, NCH, MASKED>::store(c_ptr, cur, l32, width);
    }
    ++w_words;
  }
};

template <typename T, int NCH, bool ATOMIC, bool MASKED, bool PIPE = false>
__global__ __launch_bounds__(64 * kWavesPerBlock) void update_slice2_kernel(
    KernelArgs a, const int64_t* __restrict__ pair_offsets,
    const float* __restrict__ f_in, const float* __restrict__ f_loc,
    float world_scale) {
  __shared__ int32_t sbuf[kWavesPerBlock][kMaxSent];
  __shared__ uint32_t tbuf[kWavesPerBlock][64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  UpdateSlicePhase2<T, NCH, ATOMIC, MASKED, PIPE> ph{};
  ph.syn0 = (T*)a.syn0;
  ph.syn1 = (T*)a.syn1;
  ph.stride = a.stride;
  ph.alpha = a.alpha;
  ph.l32 = lane & 31;
  ph.half = lane >> 5;
  ph.width = a.width;
  ph.world_scale = world_scale;
  ph.atomic_below = a.atomic_below;
  ph.atomic_floor = a.atomic_floor;
  ph.exp_table = a.exp_table;
  ph.exp_table_size = a.exp_table_size;
  for (int64_t s = wave_gid; s < a.num_sentences; s += total_waves) {
    ph.f_base = f_in + pair_offsets[s];
    ph.f_loc = f_loc ? f_loc + pair_offsets[s] : nullptr;
    const uint64_t base = sent_base(a.seed, (uint64_t)(a.sent_id_base + s));
    walk_sentence_dev(a.tokens, a.offsets[s],
                      (int)(a.offsets[s + 1] - a.offsets[s]), base, a.keep_thr,
                      a.table, a.table_size, a.window, a.n_neg, a.ref_window,
                      a.shared_neg, lane, sbuf[wave], tbuf[wave], ph);
  }
  unsigned int p2 = ph.w_pairs + __shfl_xor(ph.w_pairs, 32, 64);
  unsigned int o2 = ph.w_pos + __shfl_xor(ph.w_pos, 32, 64);
  float f2 = ph.w_fplus + __shfl_xor(ph.w_fplus, 32, 64);
  if (lane == 0 && a.d_pairs) {
    atomicAdd(a.d_pairs, (unsigned long long)p2);
    atomicAdd(a.d_positives, (unsigned long long)o2);
    atomicAdd(a.d_words, (unsigned long long)ph.w_words);
    atomicAdd(a.d_sum_fplus, (double)f2);
  }
}

// ---- pairs trainer (row-sharded engine): grouped explicit plan against
// local f32 caches; hogwild across groups with fp32 atomics on the shared
// cache rows.  Group = one center position; targets contiguous. ------------
template <int NC>
__global__ __launch_bounds__(64 * kWavesPerBlock) void train_pairs_kernel(
    float* __restrict__ cache0, float* __restrict__ cache1, int64_t stride,
    const int32_t* __restrict__ group_center,
    const int64_t* __restrict__ group_offsets, int64_t num_groups,
    const int32_t* __restrict__ pair_target,
    const float* __restrict__ pair_label, float alpha,
    unsigned long long* d_pairs, unsigned long long* d_positives,
    unsigned long long* d_words, double* d_sum_fplus) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  unsigned long long w_pairs = 0, w_pos = 0, w_words = 0;
  float w_fplus = 0.0f;
  for (int64_t g = wave_gid; g < num_groups; g += total_waves) {
    float* c_ptr = cache0 + (int64_t)group_center[g] * stride;
    float c_row[NC], grad[NC];
    RowIO<float, NC>::load(c_ptr, c_row, lane);
#pragma unroll
    for (int k = 0; k < NC; ++k) grad[k] = 0.0f;
    for (int64_t p = group_offsets[g]; p < group_offsets[g + 1]; ++p) {
      float* t_ptr = cache1 + (int64_t)pair_target[p] * stride;
      float t_row[NC];
      RowIO<float, NC>::load(t_ptr, t_row, lane);
      float f = 0.0f;
#pragma unroll
      for (int k = 0; k < NC; ++k) f += c_row[k] * t_row[k];
#pragma unroll
      for (int sh = 32; sh > 0; sh >>= 1) f += __shfl_xor(f, sh, 64);
      const float label = pair_label[p];
      const float gg = (label - sigmoid_clipped(f)) * alpha;
      float delta[NC];
#pragma unroll
      for (int k = 0; k < NC; ++k) {
        grad[k] += gg * t_row[k];
        delta[k] = gg * c_row[k];
      }
      RowIO<float, NC>::atomic_add(t_ptr, delta, lane);
      ++w_pairs;
      if (label > 0.5f) {
        ++w_pos;
        w_fplus += f;
      }
    }
    RowIO<float, NC>::atomic_add(c_ptr, grad, lane);
    ++w_words;
  }
  if (lane == 0 && d_pairs) {
    atomicAdd(d_pairs, w_pairs);
    atomicAdd(d_positives, w_pos);
    atomicAdd(d_words, w_words);
    atomicAdd(d_sum_fplus, (double)w_fplus);
  }
}

// Two pairs per wave (32-lane halves, same structure as TrainPhase2) for
// the pairs trainer: 2x rows in flight per wave at the same instruction
// count.  ATOMIC=false is the hogwild variant (plain RMW on the cache,
// same update class as the fused kernel's default); serial parity keeps
// the one-pair kernel above.  T = float (world>1 delta caches) or
// uint16_t/bf16 (world-1 native-dtype cache: half the bytes, the fused
// kernel's precision class).
// atomic_below (ATOMIC only): row ids < atomic_below take the atomic path,
// the rest plain RMW — the hybrid update mode.  Meaningful when ids are
// global words (direct mode: vocab sorted by count, ids < K = Zipf head);
// cache modes pass INT32_MAX (all atomic).
template <typename T, int NCH, bool ATOMIC, bool PIPE = false>
__global__ __launch_bounds__(64 * kWavesPerBlock) void train_pairs2_kernel(
    T* __restrict__ cache0, T* __restrict__ cache1, int64_t stride,
    const int32_t* __restrict__ group_center,
    const int64_t* __restrict__ group_offsets, int64_t num_groups,
    const int32_t* __restrict__ pair_target,
    const float* __restrict__ pair_label, float alpha, int32_t atomic_below,
    int32_t atomic_floor,
    unsigned long long* d_pairs, unsigned long long* d_positives,
    unsigned long long* d_words, double* d_sum_fplus) {
  const int lane = threadIdx.x & 63;
  const int l32 = lane & 31;
  const int half = lane >> 5;
  const int wave = threadIdx.x >> 6;
  const int waves_in_block = blockDim.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * waves_in_block + wave;
  const int64_t total_waves = (int64_t)gridDim.x * waves_in_block;
  unsigned int w_pairs = 0, w_pos = 0;
  unsigned long long w_words = 0;
  float w_fplus = 0.0f;
  for (int64_t g = wave_gid; g < num_groups; g += total_waves) {
    T* c_ptr = cache0 + (int64_t)group_center[g] * stride;
    float c_row[NCH], grad[NCH];
    RowIO32<T, NCH>::load(c_ptr, c_row, l32);
#pragma unroll
    for (int k = 0; k < NCH; ++k) grad[k] = 0.0f;
    const int64_t ps = group_offsets[g];
    const int64_t pe = group_offsets[g + 1];
    auto load_blk = [&](int64_t p, int32_t& tid, T*& t_ptr,
                        float (&t_row)[NCH], bool& act, float& label) {
      const int64_t my = p + half;
      act = my < pe;
      const int64_t pid = act ? my : p;
      tid = pair_target[pid];
      label = pair_label[pid];
      t_ptr = cache1 + (int64_t)tid * stride;
      RowIO32<T, NCH>::load(t_ptr, t_row, l32);
    };
    auto do_blk = [&](int32_t tid, T* t_ptr, float (&t_row)[NCH], bool act,
                      float label) {
      float f = 0.0f;
#pragma unroll
      for (int k = 0; k < NCH; ++k) f += c_row[k] * t_row[k];
      f = half_sum_f32(f);
      const float g0 = (label - sigmoid_clipped(f)) * alpha;
      const float gg = act ? g0 : 0.0f;      // idle half: zero contribution
#pragma unroll
      for (int k = 0; k < NCH; ++k) grad[k] += gg * t_row[k];
      if (act) {
        if (ATOMIC && tid < atomic_below && tid >= atomic_floor) {
#pragma unroll
          for (int k = 0; k < NCH; ++k) t_row[k] = gg * c_row[k];
          RowIO32<T, NCH>::atomic_add(t_ptr, t_row, l32);
        } else {
#pragma unroll
          for (int k = 0; k < NCH; ++k) t_row[k] += gg * c_row[k];
          RowIO32<T, NCH>::store(t_ptr, t_row, l32);
        }
        ++w_pairs;
        if (label > 0.5f) {
          ++w_pos;
          w_fplus += f;
        }
      }
    };
    if (!PIPE) {
      for (int64_t p = ps; p < pe; p += 2) {
        int32_t tid; T* t_ptr; float t_row[NCH]; bool act; float lb;
        load_blk(p, tid, t_ptr, t_row, act, lb);
        do_blk(tid, t_ptr, t_row, act, lb);
      }
    } else if (ps < pe) {
      // 2-deep block pipeline: next block's loads issue before this
      // block's update traffic (same rationale as TrainPhase2 PIPE)
      int32_t tid0, tid1; T *p0, *p1; float b0[NCH], b1[NCH];
      bool a0, a1; float l0, l1;
      load_blk(ps, tid0, p0, b0, a0, l0);
      for (int64_t k = ps;; k += 4) {
        if (k + 2 < pe) load_blk(k + 2, tid1, p1, b1, a1, l1);
        do_blk(tid0, p0, b0, a0, l0);
        if (k + 2 >= pe) break;
        if (k + 4 < pe) load_blk(k + 4, tid0, p0, b0, a0, l0);
        do_blk(tid1, p1, b1, a1, l1);
        if (k + 4 >= pe) break;
      }
    }
#pragma unroll
    for (int k = 0; k < NCH; ++k) grad[k] += __shfl_xor(grad[k], 32, 64);
    if (ATOMIC && group_center[g] < atomic_below &&
        group_center[g] >= atomic_floor) {
      if (half == 0) RowIO32<T, NCH>::atomic_add(c_ptr, grad, l32);
    } else if (half == 0) {
      float cur[NCH];
      RowIO32<T, NCH>::load(c_ptr, cur, l32);
#pragma unroll
      for (int k = 0; k < NCH; ++k) cur[k] += grad[k];
      RowIO32<T, NCH>::store(c_ptr, cur, l32);
    }
    ++w_words;
  }
  const unsigned int p2 = w_pairs + __shfl_xor(w_pairs, 32, 64);
  const unsigned int o2 = w_pos + __shfl_xor(w_pos, 32, 64);
  const float f2 = w_fplus + __shfl_xor(w_fplus, 32, 64);
  if (lane == 0 && d_pairs) {
    atomicAdd(d_pairs, (unsigned long long)p2);
    atomicAdd(d_positives, (unsigned long long)o2);
    atomicAdd(d_words, w_words);
    atomicAdd(d_sum_fplus, (double)f2);
  }
}

// ---------------------------------------------------------------------------
// pullAverage: per-sentence mean of syn0 rows (Glint pullAverage, ml:453).
// One wave per sentence; output f32 [num_sentences][dim_out<=stride].
// ---------------------------------------------------------------------------
template <typename T, int NC>
__global__ __launch_bounds__(256) void pull_average_kernel(
    const T* __restrict__ syn0, const int32_t* __restrict__ tokens,
    const int32_t* __restrict__ offsets, int64_t num_sentences,
    int64_t stride, float* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * kWavesPerBlock + wave;
  const int64_t total_waves = (int64_t)gridDim.x * kWavesPerBlock;
  for (int64_t s = wave_gid; s < num_sentences; s += total_waves) {
    const int64_t off = offsets[s];
    const int len = (int)(offsets[s + 1] - off);
    float acc[NC];
#pragma unroll
    for (int k = 0; k < NC; ++k) acc[k] = 0.0f;
    for (int p = 0; p < len; ++p) {
      const T* row = syn0 + (int64_t)tokens[off + p] * stride;
      float v[NC];
      RowIO<T, NC>::load(row, v, lane);
#pragma unroll
      for (int k = 0; k < NC; ++k) acc[k] += v[k];
    }
    const float inv = len > 0 ? 1.0f / (float)len : 0.0f;
#pragma unroll
    for (int k = 0; k < NC; ++k) acc[k] *= inv;
    RowIO<float, NC>::store(out + s * stride, acc, lane);
  }
}

// ---------------------------------------------------------------------------
// Row gather / scatter-add / subtract — the row-sharded engine's pull/push
// hot ops (Glint pull / adjust), fused so the alltoallv wire
// buffers are built in ONE pass at the shard's native dtype: no f32
// inflation, no zero-fill pass, no torch temporaries.  Two rows per wave
// (32-lane halves), dwordx2-coalesced via RowIO32.
// ---------------------------------------------------------------------------
template <typename T, int NCH>
__global__ __launch_bounds__(256) void gather_rows_kernel(
    const T* __restrict__ src, int64_t stride,
    const int32_t* __restrict__ ids, int64_t n, T* __restrict__ dst) {
  const int lane = threadIdx.x & 63;
  const int l32 = lane & 31;
  const int half = lane >> 5;
  const int64_t wave_gid =
      (((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6);
  const int64_t total_hw = (((int64_t)gridDim.x * blockDim.x) >> 6) * 2;
  for (int64_t i = 2 * wave_gid + half; i < n; i += total_hw) {
    float v[NCH];
    RowIO32<T, NCH>::load(src + (int64_t)ids[i] * stride, v, l32);
    RowIO32<T, NCH>::store(dst + i * stride, v, l32);
  }
}

// dst[ids[i]] += src[i]; atomics (fp32 atomicAdd / gfx950 packed-bf16) so
// duplicate ids — the same shard row pulled by several ranks — sum exactly.
template <typename T, int NCH>
__global__ __launch_bounds__(256) void scatter_add_rows_kernel(
    T* __restrict__ dst, int64_t stride, const int32_t* __restrict__ ids,
    int64_t n, const T* __restrict__ src) {
  const int lane = threadIdx.x & 63;
  const int l32 = lane & 31;
  const int half = lane >> 5;
  const int64_t wave_gid =
      (((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6);
  const int64_t total_hw = (((int64_t)gridDim.x * blockDim.x) >> 6) * 2;
  for (int64_t i = 2 * wave_gid + half; i < n; i += total_hw) {
    float v[NCH];
    RowIO32<T, NCH>::load(src + i * stride, v, l32);
    RowIO32<T, NCH>::atomic_add(dst + (int64_t)ids[i] * stride, v, l32);
  }
}

// out = a - b, flat elementwise over n2 element-PAIRS (strides are
// multiples of 64 so totals are even); f32 math, one pass (the delta
// computation for the push — torch would spill two f32 temporaries).
__global__ __launch_bounds__(256) void sub_bf16_kernel(
    const uint16_t* __restrict__ a, const uint16_t* __restrict__ b,
    uint16_t* __restrict__ out, int64_t n2) {
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t tot = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = tid; i < n2; i += tot) {
    const uint32_t pa = *reinterpret_cast<const uint32_t*>(a + 2 * i);
    const uint32_t pb = *reinterpret_cast<const uint32_t*>(b + 2 * i);
    v2bf16 d;
    d[0] = (__bf16)(bf16_to_f32((uint16_t)(pa & 0xFFFF)) -
                    bf16_to_f32((uint16_t)(pb & 0xFFFF)));
    d[1] = (__bf16)(bf16_to_f32((uint16_t)(pa >> 16)) -
                    bf16_to_f32((uint16_t)(pb >> 16)));
    *reinterpret_cast<v2bf16*>(out + 2 * i) = d;
  }
}

__global__ __launch_bounds__(256) void sub_f32_kernel(
    const float* __restrict__ a, const float* __restrict__ b,
    float* __restrict__ out, int64_t n2) {
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t tot = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = tid; i < n2; i += tot) {
    const float2 pa = *reinterpret_cast<const float2*>(a + 2 * i);
    const float2 pb = *reinterpret_cast<const float2*>(b + 2 * i);
    *reinterpret_cast<float2*>(out + 2 * i) =
        make_float2(pa.x - pb.x, pa.y - pb.y);
  }
}

// ---------------------------------------------------------------------------
// scores: out[r] = dot(syn0[r], q) [/ norms[r]] — the findSynonyms GEMV
// (Glint multiply, mllib:598).  rocBLAS bf16 GEMV sustains only ~1.6 TB/s
// on this shape; a wave-per-row kernel with the query held in registers
// streams the table at HBM rate.  norms == nullptr skips the divide.
// ---------------------------------------------------------------------------
template <typename T, int NC>
__global__ __launch_bounds__(256) void scores_kernel(
    const T* __restrict__ syn0, int64_t vocab, int64_t stride,
    const float* __restrict__ q, const float* __restrict__ norms,
    float* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * kWavesPerBlock + wave;
  const int64_t total_waves = (int64_t)gridDim.x * kWavesPerBlock;
  float qv[NC];
  RowIO<float, NC>::load(q, qv, lane);   // query resident in registers
  for (int64_t r = wave_gid; r < vocab; r += total_waves) {
    float v[NC];
    RowIO<T, NC>::load(syn0 + r * stride, v, lane);
    float f = 0.0f;
#pragma unroll
    for (int k = 0; k < NC; ++k) f += qv[k] * v[k];
    f = wave_sum_f32(f);
    if (lane == 0) out[r] = norms ? f / norms[r] : f;
  }
}

// ---------------------------------------------------------------------------
// norms: Euclidean norm of every row (Glint norms, mllib:486).
// ---------------------------------------------------------------------------
template <typename T, int NC>
__global__ __launch_bounds__(256) void norms_kernel(
    const T* __restrict__ syn0, int64_t vocab, int64_t stride,
    float* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int64_t wave_gid = (int64_t)blockIdx.x * kWavesPerBlock + wave;
  const int64_t total_waves = (int64_t)gridDim.x * kWavesPerBlock;
  for (int64_t r = wave_gid; r < vocab; r += total_waves) {
    float v[NC];
    RowIO<T, NC>::load(syn0 + r * stride, v, lane);
    float ss = 0.0f;
#pragma unroll
    for (int k = 0; k < NC; ++k) ss += v[k] * v[k];
    ss = wave_sum_f32(ss);
    if (lane == 0) out[r] = sqrtf(ss);
  }
}

// ---------------------------------------------------------------------------
// Host-side dispatch
// ---------------------------------------------------------------------------
template <typename T, int NC>
static void launch_train_nc(const KernelArgs& a, bool atomic, int blocks,
                            int pos_blocks, int threads, hipStream_t stream) {
  if (atomic)
    hipLaunchKernelGGL((sgns_train_kernel<T, NC, true>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
  else
    hipLaunchKernelGGL((sgns_train_kernel<T, NC, false>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
}

// Supported NC values (stride = 64*NC).  The wrapper rounds the row stride
// up to the nearest supported NC.
#define FOR_EACH_NC(X) \
  X(1) X(2) X(3) X(4) X(5) X(6) X(8) X(10) X(12) X(16) X(20) X(24) X(32)

template <typename T, int NCH>
static void launch_train2_nch(const KernelArgs& a, bool atomic, int blocks,
                              int pos_blocks, int threads,
                              hipStream_t stream, bool pipe) {
  if (atomic && pipe)
    hipLaunchKernelGGL((sgns_train2_kernel<T, NCH, true, true>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
  else if (atomic)
    hipLaunchKernelGGL((sgns_train2_kernel<T, NCH, true, false>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
  else if (pipe)
    hipLaunchKernelGGL((sgns_train2_kernel<T, NCH, false, true>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
  else
    hipLaunchKernelGGL((sgns_train2_kernel<T, NCH, false, false>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
}

template <typename T>
static void launch_train2(const KernelArgs& a, int nc, bool atomic, int blocks,
                          int pos_blocks, int threads, hipStream_t stream,
                          bool pipe = false) {
  switch (nc) {
#define CASE_NC2(N)                                                          \
  case N:                                                                    \
    launch_train2_nch<T, 2 * N>(a, atomic, blocks, pos_blocks, threads,      \
                                stream, pipe);                               \
    return;
    FOR_EACH_NC(CASE_NC2)
#undef CASE_NC2
    default:
      throw std::runtime_error("unsupported NC=" + std::to_string(nc));
  }
}

template <typename T, int NCQ>
static void launch_train4_ncq(const KernelArgs& a, bool atomic, int blocks,
                              int pos_blocks, int threads,
                              hipStream_t stream) {
  if (atomic)
    hipLaunchKernelGGL((sgns_train4_kernel<T, NCQ, true>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
  else
    hipLaunchKernelGGL((sgns_train4_kernel<T, NCQ, false>),
                       dim3(blocks, pos_blocks), dim3(threads), 0, stream, a);
}

template <typename T>
static void launch_train4(const KernelArgs& a, int nc, bool atomic, int blocks,
                          int pos_blocks, int threads, hipStream_t stream) {
  switch (nc) {
#define CASE_NC4(N)                                                          \
  case N:                                                                    \
    launch_train4_ncq<T, 4 * N>(a, atomic, blocks, pos_blocks, threads,      \
                                stream);                                     \
    return;
    FOR_EACH_NC(CASE_NC4)
#undef CASE_NC4
    default:
      throw std::runtime_error("unsupported NC=" + std::to_string(nc));
  }
}

template <typename T>
static void launch_train(const KernelArgs& a, int nc, bool atomic, int blocks,
                         int pos_blocks, int threads, hipStream_t stream) {
  switch (nc) {
#define CASE_NC(N)                                                        \
  case N:                                                                 \
    launch_train_nc<T, N>(a, atomic, blocks, pos_blocks, threads, stream);\
    return;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC=" + std::to_string(nc));
  }
}

static int supported_nc(int nc_min) {
  static const int ncs[] = {1, 2, 3, 4, 5, 6, 8, 10, 12, 16, 20, 24, 32};
  for (int nc : ncs)
    if (nc >= nc_min) return nc;
  return -1;
}

static void sgns_train(uintptr_t syn0, uintptr_t syn1, int is_bf16,
                       int64_t stride, uintptr_t tokens, uintptr_t offsets,
                       int64_t num_sentences, uintptr_t keep_thr,
                       uintptr_t table, int64_t table_size, double alpha,
                       int window, int n_neg, uint64_t seed,
                       int64_t sent_id_base, int ref_window,
                       int64_t atomic_below,
                       uintptr_t stats, int blocks, int pos_blocks,
                       int threads, uintptr_t stream_ptr,
                       uintptr_t exp_table, int exp_table_size,
                       int pair2, int64_t atomic_floor, int shared_neg) {
  HIP_CLEAR_ERROR();
  if (threads != 64 && threads != 256)
    throw std::runtime_error("threads must be 64 (serial) or 256");
  if (stride % 64 != 0) throw std::runtime_error("stride must be a multiple of 64");
  const int nc = (int)(stride / 64);
  if (supported_nc(nc) != nc) throw std::runtime_error("stride/64 not a supported NC");
  if (table_size <= 0 || table_size > 0xFFFFFFFFLL)
    throw std::runtime_error("table_size out of range");
  KernelArgs a{};
  a.syn0 = (void*)syn0;
  a.syn1 = (void*)syn1;
  a.tokens = (const int32_t*)tokens;
  a.offsets = (const int32_t*)offsets;
  a.num_sentences = num_sentences;
  a.keep_thr = (const uint32_t*)keep_thr;
  a.table = (const int32_t*)table;
  a.table_size = (uint32_t)table_size;
  a.alpha = (float)alpha;
  a.window = window;
  a.n_neg = n_neg;
  a.seed = seed;
  a.sent_id_base = sent_id_base;
  a.stride = stride;
  a.ref_window = ref_window;
  a.atomic_below = (int32_t)std::min<int64_t>(atomic_below, 0x7FFFFFFFLL);
  a.atomic_floor = (int32_t)std::min<int64_t>(atomic_floor, 0x7FFFFFFFLL);
  a.shared_neg = shared_neg;
  a.exp_table = (const float*)exp_table;
  a.exp_table_size = exp_table_size;
  unsigned long long* st = (unsigned long long*)stats;
  if (st) {
    a.d_pairs = st + 0;
    a.d_positives = st + 1;
    a.d_words = st + 2;
    a.d_sum_fplus = (double*)(st + 3);
  }
  hipStream_t stream = (hipStream_t)stream_ptr;
  const bool use_atomic = atomic_below != 0;
  if (pos_blocks < 1) pos_blocks = 1;
  if (pair2 == 3 && threads == 256) {
    if (is_bf16)
      launch_train2<uint16_t>(a, nc, use_atomic, blocks, pos_blocks, threads,
                              stream, true);
    else
      launch_train2<float>(a, nc, use_atomic, blocks, pos_blocks, threads,
                           stream, true);
  } else if (pair2 == 2 && threads == 256) {
    if (is_bf16)
      launch_train4<uint16_t>(a, nc, use_atomic, blocks, pos_blocks, threads,
                              stream);
    else
      launch_train4<float>(a, nc, use_atomic, blocks, pos_blocks, threads,
                           stream);
  } else if (pair2 && threads == 256) {
    if (is_bf16)
      launch_train2<uint16_t>(a, nc, use_atomic, blocks, pos_blocks, threads,
                              stream);
    else
      launch_train2<float>(a, nc, use_atomic, blocks, pos_blocks, threads,
                           stream);
  } else if (is_bf16) {
    launch_train<uint16_t>(a, nc, use_atomic, blocks, pos_blocks, threads,
                           stream);
  } else {
    launch_train<float>(a, nc, use_atomic, blocks, pos_blocks, threads,
                        stream);
  }
  HIP_CHECK(hipGetLastError());
}

static KernelArgs make_walk_args(uintptr_t syn0, uintptr_t syn1,
                                 int64_t stride, uintptr_t tokens,
                                 uintptr_t offsets, int64_t num_sentences,
                                 uintptr_t keep_thr, uintptr_t table,
                                 int64_t table_size, double alpha, int window,
                                 int n_neg, uint64_t seed,
                                 int64_t sent_id_base, int ref_window,
                                 uintptr_t stats) {
  if (stride % 8 != 0)
    throw std::runtime_error("stride must be a multiple of 8");
  if (table_size <= 0 || table_size > 0xFFFFFFFFLL)
    throw std::runtime_error("table_size out of range");
  KernelArgs a{};
  a.syn0 = (void*)syn0;
  a.syn1 = (void*)syn1;
  a.tokens = (const int32_t*)tokens;
  a.offsets = (const int32_t*)offsets;
  a.num_sentences = num_sentences;
  a.keep_thr = (const uint32_t*)keep_thr;
  a.table = (const int32_t*)table;
  a.table_size = (uint32_t)table_size;
  a.alpha = (float)alpha;
  a.window = window;
  a.n_neg = n_neg;
  a.seed = seed;
  a.sent_id_base = sent_id_base;
  a.stride = stride;
  a.ref_window = ref_window;
  unsigned long long* st = (unsigned long long*)stats;
  if (st) {
    a.d_pairs = st + 0;
    a.d_positives = st + 1;
    a.d_words = st + 2;
    a.d_sum_fplus = (double*)(st + 3);
  }
  return a;
}

static void count_pairs(uintptr_t tokens, uintptr_t offsets,
                        int64_t num_sentences, uintptr_t keep_thr,
                        uintptr_t table, int64_t table_size, int window,
                        int n_neg, uint64_t seed, int64_t sent_id_base,
                        int ref_window, uintptr_t counts_out, int blocks,
                        int threads, uintptr_t stream_ptr,
                        int shared_neg) {
  HIP_CLEAR_ERROR();
  KernelArgs a = make_walk_args(0, 0, 64, tokens, offsets, num_sentences,
                                keep_thr, table, table_size, 0.0, window,
                                n_neg, seed, sent_id_base, ref_window, 0);
  a.shared_neg = shared_neg;
  hipLaunchKernelGGL(count_pairs_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream_ptr, a, (int64_t*)counts_out);
  HIP_CHECK(hipGetLastError());
}

static void plan_emit(uintptr_t tokens, uintptr_t offsets,
                      int64_t num_sentences, uintptr_t keep_thr,
                      uintptr_t table, int64_t table_size, int window,
                      int n_neg, uint64_t seed, int64_t sent_id_base,
                      int ref_window, uintptr_t pair_offsets,
                      uintptr_t out_target, uintptr_t out_label,
                      uintptr_t out_start, uintptr_t out_center, int blocks,
                      int threads, uintptr_t stream_ptr, int shared_neg) {
  HIP_CLEAR_ERROR();
  KernelArgs a = make_walk_args(0, 0, 64, tokens, offsets, num_sentences,
                                keep_thr, table, table_size, 0.0, window,
                                n_neg, seed, sent_id_base, ref_window, 0);
  a.shared_neg = shared_neg;
  hipLaunchKernelGGL(plan_emit_kernel, dim3(blocks), dim3(threads), 0,
                     (hipStream_t)stream_ptr, a,
                     (const int64_t*)pair_offsets, (int32_t*)out_target,
                     (float*)out_label, (uint8_t*)out_start,
                     (int32_t*)out_center);
  HIP_CHECK(hipGetLastError());
}

static void dots_slice(uintptr_t syn0, uintptr_t syn1, int is_bf16,
                       int64_t stride, uintptr_t tokens, uintptr_t offsets,
                       int64_t num_sentences, uintptr_t keep_thr,
                       uintptr_t table, int64_t table_size, int window,
                       int n_neg, uint64_t seed, int64_t sent_id_base,
                       int ref_window, uintptr_t pair_offsets, uintptr_t f_out,
                       int blocks, int threads, uintptr_t stream_ptr,
                       int pair_mode, int width, int shared_neg) {
  HIP_CLEAR_ERROR();
  if (width <= 0) width = (int)stride;
  KernelArgs a = make_walk_args(syn0, syn1, stride, tokens, offsets,
                                num_sentences, keep_thr, table, table_size,
                                0.0, window, n_neg, seed, sent_id_base,
                                ref_window, 0);
  a.width = width;
  a.shared_neg = shared_neg;
  const int nc = supported_nc((width + 63) / 64);
  if (width != (int)stride && !(pair_mode && threads == 256))
    throw std::runtime_error("masked width requires pair_mode kernels");
  hipStream_t stream = (hipStream_t)stream_ptr;
  const bool masked = width != (int)stride;
#define DOTS_CASE(T, N)                                                       \
  do {                                                                        \
    if (pair_mode && threads == 256 && masked)                                \
      hipLaunchKernelGGL((dots_slice2_kernel<T, 2 * N, true>), dim3(blocks),  \
                         dim3(threads), 0, stream, a,                         \
                         (const int64_t*)pair_offsets, (float*)f_out);        \
    else if (pair_mode && threads == 256)                                     \
      hipLaunchKernelGGL((dots_slice2_kernel<T, 2 * N, false>), dim3(blocks), \
                         dim3(threads), 0, stream, a,                         \
                         (const int64_t*)pair_offsets, (float*)f_out);        \
    else                                                                      \
      hipLaunchKernelGGL((dots_slice_kernel<T, N>), dim3(blocks),             \
                         dim3(threads), 0, stream, a,                         \
                         (const int64_t*)pair_offsets, (float*)f_out);        \
  } while (0)
  switch (nc) {
#define CASE_NC(N)                                   \
  case N:                                            \
    if (is_bf16) { DOTS_CASE(uint16_t, N); }         \
    else { DOTS_CASE(float, N); }                    \
    break;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC");
  }
#undef DOTS_CASE
  HIP_CHECK(hipGetLastError());
}

template <typename T, int NCH>
static void launch_update2_k(const KernelArgs& a, bool atomic, bool masked,
                             bool pipe, int blocks, int threads,
                             hipStream_t stream, const int64_t* poff,
                             const float* fin, const float* floc, float ws) {
#define UPD2(AT, MS, PP)                                                   \
  hipLaunchKernelGGL((update_slice2_kernel<T, NCH, AT, MS, PP>),           \
                     dim3(blocks), dim3(threads), 0, stream, a, poff, fin, \
                     floc, ws)
  if (atomic) {
    if (masked) {
      if (pipe) UPD2(true, true, true); else UPD2(true, true, false);
    } else {
      if (pipe) UPD2(true, false, true); else UPD2(true, false, false);
    }
  } else {
    if (masked) {
      if (pipe) UPD2(false, true, true); else UPD2(false, true, false);
    } else {
      if (pipe) UPD2(false, false, true); else UPD2(false, false, false);
    }
  }
#undef UPD2
}

static void update_slice(uintptr_t syn0, uintptr_t syn1, int is_bf16,
                         int64_t stride, uintptr_t tokens, uintptr_t offsets,
                         int64_t num_sentences, uintptr_t keep_thr,
                         uintptr_t table, int64_t table_size, double alpha,
                         int window, int n_neg, uint64_t seed,
                         int64_t sent_id_base, int ref_window,
                         uintptr_t pair_offsets, uintptr_t f_in,
                         uintptr_t f_loc, double world_scale,
                         int64_t atomic_below,
                         uintptr_t stats, int blocks, int threads,
                         uintptr_t stream_ptr,
                         uintptr_t exp_table, int exp_table_size,
                         int pair_mode, int width, int64_t atomic_floor,
                         int shared_neg) {
  HIP_CLEAR_ERROR();
  if (width <= 0) width = (int)stride;
  KernelArgs a = make_walk_args(syn0, syn1, stride, tokens, offsets,
                                num_sentences, keep_thr, table, table_size,
                                alpha, window, n_neg, seed, sent_id_base,
                                ref_window, stats);
  a.width = width;
  a.shared_neg = shared_neg;
  a.atomic_below = (int32_t)std::min<int64_t>(atomic_below, 0x7FFFFFFFLL);
  a.atomic_floor = (int32_t)std::min<int64_t>(atomic_floor, 0x7FFFFFFFLL);
  a.exp_table = (const float*)exp_table;
  a.exp_table_size = exp_table_size;
  const int atomic = atomic_below > 0;
  const int nc = supported_nc((width + 63) / 64);
  if (width != (int)stride && !(pair_mode && threads == 256))
    throw std::runtime_error("masked width requires pair_mode kernels");
  hipStream_t stream = (hipStream_t)stream_ptr;
  const bool masked = width != (int)stride;
#define UPD_CASE(T, N)                                                        \
  do {                                                                        \
    if (pair_mode && threads == 256) {                                        \
      launch_update2_k<T, 2 * N>(a, atomic, masked, pair_mode == 3, blocks,   \
                                 threads, stream,                             \
                                 (const int64_t*)pair_offsets,                \
                                 (const float*)f_in, (const float*)f_loc,     \
                                 (float)world_scale);                         \
    } else if (atomic) {                                                      \
      hipLaunchKernelGGL((update_slice_kernel<T, N, true>), dim3(blocks),     \
                         dim3(threads), 0, stream, a,                         \
                         (const int64_t*)pair_offsets, (const float*)f_in,    \
                         (const float*)f_loc, (float)world_scale);            \
    } else {                                                                  \
      hipLaunchKernelGGL((update_slice_kernel<T, N, false>), dim3(blocks),    \
                         dim3(threads), 0, stream, a,                         \
                         (const int64_t*)pair_offsets, (const float*)f_in,    \
                         (const float*)f_loc, (float)world_scale);            \
    }                                                                         \
  } while (0)
  switch (nc) {
#define CASE_NC(N)                                  \
  case N:                                           \
    if (is_bf16) { UPD_CASE(uint16_t, N); }         \
    else { UPD_CASE(float, N); }                    \
    break;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC");
  }
#undef UPD_CASE
  HIP_CHECK(hipGetLastError());
}

static void train_pairs(uintptr_t cache0, uintptr_t cache1, int64_t stride,
                        uintptr_t group_center, uintptr_t group_offsets,
                        int64_t num_groups, uintptr_t pair_target,
                        uintptr_t pair_label, double alpha, uintptr_t stats,
                        int blocks, int threads, uintptr_t stream_ptr,
                        int pair_mode, int atomic, int is_bf16,
                        int64_t atomic_below, int64_t atomic_floor) {
  HIP_CLEAR_ERROR();
  if (stride % 64 != 0) throw std::runtime_error("stride must be a multiple of 64");
  if (is_bf16 && pair_mode == 0)
    throw std::runtime_error("bf16 pairs cache requires pair_mode=1");
  const int nc = (int)(stride / 64);
  const int32_t abelow =
      (int32_t)std::min<int64_t>(atomic_below, 0x7FFFFFFFLL);
  const int32_t afloor =
      (int32_t)std::min<int64_t>(atomic_floor, 0x7FFFFFFFLL);
  const bool pipe = pair_mode == 3;
  hipStream_t stream = (hipStream_t)stream_ptr;
  unsigned long long* st = (unsigned long long*)stats;
  switch (nc) {
#define CASE_NC(N)                                                            \
  case N:                                                                     \
    if (pair_mode == 0) {                                                     \
      hipLaunchKernelGGL((train_pairs_kernel<N>), dim3(blocks),               \
                         dim3(threads), 0, stream, (float*)cache0,            \
                         (float*)cache1, stride,                              \
                         (const int32_t*)group_center,                        \
                         (const int64_t*)group_offsets, num_groups,           \
                         (const int32_t*)pair_target,                         \
                         (const float*)pair_label, (float)alpha,              \
                         st ? st + 0 : nullptr, st ? st + 1 : nullptr,        \
                         st ? st + 2 : nullptr,                               \
                         st ? (double*)(st + 3) : nullptr);                   \
    } else if (is_bf16) {                                                     \
      if (atomic)                                                             \
        if (pipe)                                                             \
        hipLaunchKernelGGL((train_pairs2_kernel<uint16_t, 2 * N, true, true>), \
                           dim3(blocks), dim3(threads), 0, stream,      \
                           (uint16_t*)cache0, (uint16_t*)cache1, stride,  \
                           (const int32_t*)group_center,                \
                           (const int64_t*)group_offsets, num_groups,   \
                           (const int32_t*)pair_target,                 \
                           (const float*)pair_label, (float)alpha,      \
                           abelow, afloor,                              \
                           st ? st + 0 : nullptr, st ? st + 1 : nullptr,\
                           st ? st + 2 : nullptr,                       \
                           st ? (double*)(st + 3) : nullptr);           \
      else                                                              \
        hipLaunchKernelGGL((train_pairs2_kernel<uint16_t, 2 * N, true>),      \
                           dim3(blocks), dim3(threads), 0, stream,            \
                           (uint16_t*)cache0, (uint16_t*)cache1, stride,      \
                           (const int32_t*)group_center,                      \
                           (const int64_t*)group_offsets, num_groups,         \
                           (const int32_t*)pair_target,                       \
                           (const float*)pair_label, (float)alpha, abelow, afloor,    \
                           st ? st + 0 : nullptr, st ? st + 1 : nullptr,      \
                           st ? st + 2 : nullptr,                             \
                           st ? (double*)(st + 3) : nullptr);                 \
      else                                                                    \
        if (pipe)                                                             \
        hipLaunchKernelGGL((train_pairs2_kernel<uint16_t, 2 * N, false, true>), \
                           dim3(blocks), dim3(threads), 0, stream,      \
                           (uint16_t*)cache0, (uint16_t*)cache1, stride,  \
                           (const int32_t*)group_center,                \
                           (const int64_t*)group_offsets, num_groups,   \
                           (const int32_t*)pair_target,                 \
                           (const float*)pair_label, (float)alpha,      \
                           abelow, afloor,                              \
                           st ? st + 0 : nullptr, st ? st + 1 : nullptr,\
                           st ? st + 2 : nullptr,                       \
                           st ? (double*)(st + 3) : nullptr);           \
      else                                                              \
        hipLaunchKernelGGL((train_pairs2_kernel<uint16_t, 2 * N, false>),     \
                           dim3(blocks), dim3(threads), 0, stream,            \
                           (uint16_t*)cache0, (uint16_t*)cache1, stride,      \
                           (const int32_t*)group_center,                      \
                           (const int64_t*)group_offsets, num_groups,         \
                           (const int32_t*)pair_target,                       \
                           (const float*)pair_label, (float)alpha, abelow, afloor,    \
                           st ? st + 0 : nullptr, st ? st + 1 : nullptr,      \
                           st ? st + 2 : nullptr,                             \
                           st ? (double*)(st + 3) : nullptr);                 \
    } else if (atomic) {                                                      \
      if (pipe)                                                             \
        hipLaunchKernelGGL((train_pairs2_kernel<float, 2 * N, true, true>), \
                           dim3(blocks), dim3(threads), 0, stream,      \
                           (float*)cache0, (float*)cache1, stride,  \
                           (const int32_t*)group_center,                \
                           (const int64_t*)group_offsets, num_groups,   \
                           (const int32_t*)pair_target,                 \
                           (const float*)pair_label, (float)alpha,      \
                           abelow, afloor,                              \
                           st ? st + 0 : nullptr, st ? st + 1 : nullptr,\
                           st ? st + 2 : nullptr,                       \
                           st ? (double*)(st + 3) : nullptr);           \
      else                                                              \
        hipLaunchKernelGGL((train_pairs2_kernel<float, 2 * N, true>),           \
                         dim3(blocks), dim3(threads), 0, stream,              \
                         (float*)cache0,                                      \
                         (float*)cache1, stride,                              \
                         (const int32_t*)group_center,                        \
                         (const int64_t*)group_offsets, num_groups,           \
                         (const int32_t*)pair_target,                         \
                         (const float*)pair_label, (float)alpha, abelow, afloor,      \
                         st ? st + 0 : nullptr, st ? st + 1 : nullptr,        \
                         st ? st + 2 : nullptr,                               \
                         st ? (double*)(st + 3) : nullptr);                   \
    } else {                                                                  \
      if (pipe)                                                             \
        hipLaunchKernelGGL((train_pairs2_kernel<float, 2 * N, false, true>), \
                           dim3(blocks), dim3(threads), 0, stream,      \
                           (float*)cache0, (float*)cache1, stride,  \
                           (const int32_t*)group_center,                \
                           (const int64_t*)group_offsets, num_groups,   \
                           (const int32_t*)pair_target,                 \
                           (const float*)pair_label, (float)alpha,      \
                           abelow, afloor,                              \
                           st ? st + 0 : nullptr, st ? st + 1 : nullptr,\
                           st ? st + 2 : nullptr,                       \
                           st ? (double*)(st + 3) : nullptr);           \
      else                                                              \
        hipLaunchKernelGGL((train_pairs2_kernel<float, 2 * N, false>),          \
                         dim3(blocks), dim3(threads), 0, stream,              \
                         (float*)cache0,                                      \
                         (float*)cache1, stride,                              \
                         (const int32_t*)group_center,                        \
                         (const int64_t*)group_offsets, num_groups,           \
                         (const int32_t*)pair_target,                         \
                         (const float*)pair_label, (float)alpha, abelow, afloor,      \
                         st ? st + 0 : nullptr, st ? st + 1 : nullptr,        \
                         st ? st + 2 : nullptr,                               \
                         st ? (double*)(st + 3) : nullptr);                   \
    }                                                                         \
    break;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC");
  }
  HIP_CHECK(hipGetLastError());
}

template <typename T>
static void launch_pull_average_t(uintptr_t syn0, uintptr_t tokens,
                                  uintptr_t offsets, int64_t num_sentences,
                                  int64_t stride, uintptr_t out, int blocks,
                                  hipStream_t stream) {
  const int nc = (int)(stride / 64);
  switch (nc) {
#define CASE_NC(N)                                                          \
  case N:                                                                   \
    hipLaunchKernelGGL((pull_average_kernel<T, N>), dim3(blocks), dim3(256),\
                       0, stream, (const T*)syn0, (const int32_t*)tokens,   \
                       (const int32_t*)offsets, num_sentences, stride,      \
                       (float*)out);                                        \
    return;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC");
  }
}

static void pull_average(uintptr_t syn0, int is_bf16, int64_t stride,
                         uintptr_t tokens, uintptr_t offsets,
                         int64_t num_sentences, uintptr_t out, int blocks,
                         uintptr_t stream_ptr) {
  HIP_CLEAR_ERROR();
  hipStream_t stream = (hipStream_t)stream_ptr;
  if (is_bf16)
    launch_pull_average_t<uint16_t>(syn0, tokens, offsets, num_sentences,
                                    stride, out, blocks, stream);
  else
    launch_pull_average_t<float>(syn0, tokens, offsets, num_sentences, stride,
                                 out, blocks, stream);
  HIP_CHECK(hipGetLastError());
}

template <typename T>
static void launch_scores_t(uintptr_t syn0, int64_t vocab, int64_t stride,
                            uintptr_t q, uintptr_t norms, uintptr_t out,
                            int blocks, hipStream_t stream) {
  const int nc = (int)(stride / 64);
  switch (nc) {
#define CASE_NC(N)                                                         \
  case N:                                                                  \
    hipLaunchKernelGGL((scores_kernel<T, N>), dim3(blocks), dim3(256), 0,  \
                       stream, (const T*)syn0, vocab, stride,              \
                       (const float*)q, (const float*)norms, (float*)out); \
    return;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC");
  }
}

static void scores(uintptr_t syn0, int is_bf16, int64_t vocab,
                   int64_t stride, uintptr_t q, uintptr_t norms,
                   uintptr_t out, int blocks, uintptr_t stream_ptr) {
  HIP_CLEAR_ERROR();
  hipStream_t stream = (hipStream_t)stream_ptr;
  if (is_bf16)
    launch_scores_t<uint16_t>(syn0, vocab, stride, q, norms, out, blocks,
                              stream);
  else
    launch_scores_t<float>(syn0, vocab, stride, q, norms, out, blocks,
                           stream);
  HIP_CHECK(hipGetLastError());
}

template <typename T>
static void launch_norms_t(uintptr_t syn0, int64_t vocab, int64_t stride,
                           uintptr_t out, int blocks, hipStream_t stream) {
  const int nc = (int)(stride / 64);
  switch (nc) {
#define CASE_NC(N)                                                        \
  case N:                                                                 \
    hipLaunchKernelGGL((norms_kernel<T, N>), dim3(blocks), dim3(256), 0,  \
                       stream, (const T*)syn0, vocab, stride, (float*)out);\
    return;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC");
  }
}

static void norms(uintptr_t syn0, int is_bf16, int64_t vocab, int64_t stride,
                  uintptr_t out, int blocks, uintptr_t stream_ptr) {
  HIP_CLEAR_ERROR();
  hipStream_t stream = (hipStream_t)stream_ptr;
  if (is_bf16)
    launch_norms_t<uint16_t>(syn0, vocab, stride, out, blocks, stream);
  else
    launch_norms_t<float>(syn0, vocab, stride, out, blocks, stream);
  HIP_CHECK(hipGetLastError());
}

template <typename T>
static void launch_gather_scatter(const char* which, uintptr_t table,
                                  int64_t stride, uintptr_t ids, int64_t n,
                                  uintptr_t buf, int blocks,
                                  hipStream_t stream) {
  const int nc = (int)(stride / 64);
  switch (nc) {
#define CASE_NC(N)                                                           \
  case N:                                                                    \
    if (which[0] == 'g')                                                     \
      hipLaunchKernelGGL((gather_rows_kernel<T, 2 * N>), dim3(blocks),       \
                         dim3(256), 0, stream, (const T*)table, stride,      \
                         (const int32_t*)ids, n, (T*)buf);                   \
    else                                                                     \
      hipLaunchKernelGGL((scatter_add_rows_kernel<T, 2 * N>), dim3(blocks),  \
                         dim3(256), 0, stream, (T*)table, stride,            \
                         (const int32_t*)ids, n, (const T*)buf);             \
    return;
    FOR_EACH_NC(CASE_NC)
#undef CASE_NC
    default:
      throw std::runtime_error("unsupported NC");
  }
}

static int rows_blocks(int64_t n) {
  // 2 rows per wave, 4 waves per block -> n/8 blocks fills; cap 8192
  return (int)std::max<int64_t>(1, std::min<int64_t>((n + 7) / 8, 8192));
}

static void gather_rows(uintptr_t src, int is_bf16, int64_t stride,
                        uintptr_t ids, int64_t n, uintptr_t dst,
                        uintptr_t stream_ptr) {
  HIP_CLEAR_ERROR();
  if (stride % 64 != 0)
    throw std::runtime_error("stride must be a multiple of 64");
  if (n == 0) return;
  hipStream_t stream = (hipStream_t)stream_ptr;
  if (is_bf16)
    launch_gather_scatter<uint16_t>("g", src, stride, ids, n, dst,
                                    rows_blocks(n), stream);
  else
    launch_gather_scatter<float>("g", src, stride, ids, n, dst,
                                 rows_blocks(n), stream);
  HIP_CHECK(hipGetLastError());
}

static void scatter_add_rows(uintptr_t dst, int is_bf16, int64_t stride,
                             uintptr_t ids, int64_t n, uintptr_t src,
                             uintptr_t stream_ptr) {
  HIP_CLEAR_ERROR();
  if (stride % 64 != 0)
    throw std::runtime_error("stride must be a multiple of 64");
  if (n == 0) return;
  hipStream_t stream = (hipStream_t)stream_ptr;
  if (is_bf16)
    launch_gather_scatter<uint16_t>("s", dst, stride, ids, n, src,
                                    rows_blocks(n), stream);
  else
    launch_gather_scatter<float>("s", dst, stride, ids, n, src,
                                 rows_blocks(n), stream);
  HIP_CHECK(hipGetLastError());
}

static void sub_rows(uintptr_t a, uintptr_t b, int is_bf16,
                     int64_t total_elems, uintptr_t out,
                     uintptr_t stream_ptr) {
  HIP_CLEAR_ERROR();
  if (total_elems % 2 != 0)
    throw std::runtime_error("total_elems must be even");
  if (total_elems == 0) return;
  const int64_t n2 = total_elems / 2;
  const int blocks =
      (int)std::max<int64_t>(1, std::min<int64_t>((n2 + 255) / 256, 8192));
  hipStream_t stream = (hipStream_t)stream_ptr;
  if (is_bf16)
    hipLaunchKernelGGL(sub_bf16_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const uint16_t*)a, (const uint16_t*)b, (uint16_t*)out,
                       n2);
  else
    hipLaunchKernelGGL(sub_f32_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const float*)a, (const float*)b, (float*)out, n2);
  HIP_CHECK(hipGetLastError());
}

static int round_stride(int dim) {
  int nc = supported_nc((dim + 63) / 64);
  if (nc < 0) throw std::runtime_error("dim too large (max 2048)");
  return 64 * nc;
}

PYBIND11_MODULE(_hip_native, m) {
  m.doc() = "MI355X (gfx950) fused SGNS kernels";
  m.def("sgns_train", &sgns_train, py::arg("syn0"), py::arg("syn1"),
        py::arg("is_bf16"), py::arg("stride"), py::arg("tokens"),
        py::arg("offsets"), py::arg("num_sentences"), py::arg("keep_thr"),
        py::arg("table"), py::arg("table_size"), py::arg("alpha"),
        py::arg("window"), py::arg("n_neg"), py::arg("seed"),
        py::arg("sent_id_base"), py::arg("ref_window"), py::arg("atomic"),
        py::arg("stats"), py::arg("blocks"), py::arg("pos_blocks"),
        py::arg("threads"), py::arg("stream"),
        py::arg("exp_table") = 0, py::arg("exp_table_size") = 0,
        py::arg("pair2") = 0, py::arg("atomic_floor") = 0,
        py::arg("shared_neg") = 0);
  m.def("count_pairs", &count_pairs);
  m.def("dots_slice", &dots_slice);
  m.def("update_slice", &update_slice);
  m.def("train_pairs", &train_pairs);
  m.def("plan_emit", &plan_emit);
  m.def("pull_average", &pull_average);
  m.def("norms", &norms);
  m.def("scores", &scores);
  m.def("gather_rows", &gather_rows);
  m.def("scatter_add_rows", &scatter_add_rows);
  m.def("sub_rows", &sub_rows);
  m.def("round_stride", &round_stride);
  m.def("max_sentence_length", []() { return (int)kMaxSent; });
}
