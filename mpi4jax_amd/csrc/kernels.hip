// CDNA4 (gfx950) kernels for mpi4jax_amd: elementwise combine (scan ring)
// and LDS-staged strided pack/unpack.
//
// Replaces what in the reference is done by the host MPI library's
// reduction machinery (there are no device kernels anywhere in the
// reference — SURVEY.md §2.2).  Written for 64-wide wavefronts; blocks are
// 256 threads (4 waves) and grids are sized ≫256 workgroups where the
// problem allows so all 8 XCDs fill.

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>

#include <type_traits>

#include "kernels.h"

namespace {

constexpr int kBlock = 256;  // 4 wave64s

template <typename T>
struct AccOf {
  using type = T;
};
// do f16/bf16 math in f32
template <>
struct AccOf<__half> {
  using type = float;
};
template <>
struct AccOf<__hip_bfloat16> {
  using type = float;
};

template <typename T>
__device__ inline typename AccOf<T>::type to_acc(T v) {
  return (typename AccOf<T>::type)v;
}
template <>
__device__ inline float to_acc<__half>(__half v) {
  return __half2float(v);
}
template <>
__device__ inline float to_acc<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}

template <typename T>
__device__ inline T from_acc(typename AccOf<T>::type v) {
  return (T)v;
}
template <>
__device__ inline __half from_acc<__half>(float v) {
  return __float2half(v);
}
template <>
__device__ inline __hip_bfloat16 from_acc<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

template <typename A>
__device__ inline A apply_op(int op, A x, A y) {
  switch (op) {
    case OPC_SUM: return x + y;
    case OPC_PROD: return x * y;
    case OPC_MAX: return x > y ? x : y;
    case OPC_MIN: return x < y ? x : y;
    default:
      if constexpr (std::is_integral<A>::value) {
        switch (op) {
          case OPC_BAND: return x & y;
          case OPC_BOR: return x | y;
          default: return x ^ y;  // OPC_BXOR
        }
      } else {
        // bitwise on floating dtypes is rejected in the Python layer
        return x;
      }
  }
}

template <typename T>
__global__ void combine_kernel(T* __restrict__ dst, const T* __restrict__ a,
                               const T* __restrict__ b, long long n,
                               int op) {
  using A = typename AccOf<T>::type;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long step = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += step) {
    A x = to_acc<T>(a[i]);
    A y = to_acc<T>(b[i]);
    dst[i] = from_acc<T>(apply_op(op, x, y));
  }
}

int grid_for(long long n) {
  long long blocks = (n + kBlock - 1) / kBlock;
  // ≥8 blocks per XCD when there is work; cap so tiny n doesn't overlaunch
  if (blocks > 8192) blocks = 8192;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

template <typename T>
void launch_combine_t(void* dst, const void* a, const void* b, long long n,
                      int op, hipStream_t stream) {
  hipLaunchKernelGGL(combine_kernel<T>, dim3(grid_for(n)), dim3(kBlock), 0,
                     stream, (T*)dst, (const T*)a, (const T*)b, n, op);
}

// ------------------------------------------------------------- pack/unpack

// Tiled 2-D gather through LDS.  One 256-thread block moves a 64(row) ×
// 32(col) tile: global reads walk the *contiguous* source axis within a
// wave where possible; the LDS stage (padded to kill bank conflicts)
// decouples read order from the contiguous write order.
constexpr int TILE_R = 64;
constexpr int TILE_C = 32;

template <typename T>
__global__ void pack2d_kernel(T* __restrict__ out, const T* __restrict__ in,
                              long long rows, long long cols,
                              long long stride0, long long stride1) {
  __shared__ T tile[TILE_R][TILE_C + 1];
  long long tiles_c = (cols + TILE_C - 1) / TILE_C;
  for (long long t = blockIdx.x; ; t += gridDim.x) {
    long long tr = (t / tiles_c) * TILE_R;
    long long tc = (t % tiles_c) * TILE_C;
    if (tr >= rows) break;
    // stage: thread (r, c) lanes sweep columns fastest → coalesced when
    // stride1 == 1 (row-slices); for column-slices stride0 == 1 and the
    // read sweep below is swapped by indexing math.
    for (int i = threadIdx.x; i < TILE_R * TILE_C; i += blockDim.x) {
      int r = i / TILE_C, c = i % TILE_C;
      long long gr = tr + r, gc = tc + c;
      if (gr < rows && gc < cols) {
        tile[r][c] = in[gr * stride0 + gc * stride1];
      }
    }
    __syncthreads();
    for (int i = threadIdx.x; i < TILE_R * TILE_C; i += blockDim.x) {
      int r = i / TILE_C, c = i % TILE_C;
      long long gr = tr + r, gc = tc + c;
      if (gr < rows && gc < cols) {
        out[gr * cols + gc] = tile[r][c];
      }
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void unpack2d_kernel(T* __restrict__ out,
                                const T* __restrict__ in, long long rows,
                                long long cols, long long stride0,
                                long long stride1) {
  long long n = rows * cols;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long step = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += step) {
    long long r = i / cols, c = i % cols;
    out[r * stride0 + c * stride1] = in[i];
  }
}

template <typename T>
void launch_pack2d_t(void* out, const void* in, long long rows,
                     long long cols, long long s0, long long s1,
                     hipStream_t stream) {
  long long tiles =
      ((rows + TILE_R - 1) / TILE_R) * ((cols + TILE_C - 1) / TILE_C);
  int grid = (int)(tiles < 8192 ? (tiles < 1 ? 1 : tiles) : 8192);
  hipLaunchKernelGGL(pack2d_kernel<T>, dim3(grid), dim3(kBlock), 0, stream,
                     (T*)out, (const T*)in, rows, cols, s0, s1);
}

template <typename T>
void launch_unpack2d_t(void* out, const void* in, long long rows,
                       long long cols, long long s0, long long s1,
                       hipStream_t stream) {
  hipLaunchKernelGGL(unpack2d_kernel<T>, dim3(grid_for(rows * cols)),
                     dim3(kBlock), 0, stream, (T*)out, (const T*)in, rows,
                     cols, s0, s1);
}

}  // namespace

void launch_combine(void* dst, const void* a, const void* b, long long n,
                    int dt, int op, hipStream_t stream) {
  switch (dt) {
    case DT_F32: launch_combine_t<float>(dst, a, b, n, op, stream); break;
    case DT_F64: launch_combine_t<double>(dst, a, b, n, op, stream); break;
    case DT_F16: launch_combine_t<__half>(dst, a, b, n, op, stream); break;
    case DT_BF16:
      launch_combine_t<__hip_bfloat16>(dst, a, b, n, op, stream);
      break;
    case DT_I8: launch_combine_t<signed char>(dst, a, b, n, op, stream); break;
    case DT_U8:
      launch_combine_t<unsigned char>(dst, a, b, n, op, stream);
      break;
    case DT_I32: launch_combine_t<int>(dst, a, b, n, op, stream); break;
    case DT_I64:
      launch_combine_t<long long>(dst, a, b, n, op, stream);
      break;
  }
}

void launch_pack2d(void* out, const void* in, long long rows, long long cols,
                   long long s0, long long s1, int elem_size,
                   hipStream_t stream) {
  switch (elem_size) {
    case 1: launch_pack2d_t<unsigned char>(out, in, rows, cols, s0, s1, stream); break;
    case 2: launch_pack2d_t<unsigned short>(out, in, rows, cols, s0, s1, stream); break;
    case 4: launch_pack2d_t<unsigned int>(out, in, rows, cols, s0, s1, stream); break;
    case 8: launch_pack2d_t<unsigned long long>(out, in, rows, cols, s0, s1, stream); break;
  }
}

void launch_unpack2d(void* out, const void* in, long long rows,
                     long long cols, long long s0, long long s1,
                     int elem_size, hipStream_t stream) {
  switch (elem_size) {
    case 1: launch_unpack2d_t<unsigned char>(out, in, rows, cols, s0, s1, stream); break;
    case 2: launch_unpack2d_t<unsigned short>(out, in, rows, cols, s0, s1, stream); break;
    case 4: launch_unpack2d_t<unsigned int>(out, in, rows, cols, s0, s1, stream); break;
    case 8: launch_unpack2d_t<unsigned long long>(out, in, rows, cols, s0, s1, stream); break;
  }
}
