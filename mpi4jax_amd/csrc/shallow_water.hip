// Fused CDNA4 kernels for the shallow-water step (gfx950).
//
// The eager torch path runs ~300 small kernels per model step (measured,
// profiles/); these five fused stencil kernels + the h/u/v halo exchange
// are the whole step.  Key idea: the derived fields (fe/fn/q/ke and the
// friction gradients) never need a halo *exchange* — at an open edge their
// halo value equals the same formula evaluated on local data (h/u/v halos
// are fresh), bitwise-identical to what the neighbor would send; at a
// closed edge the eager path leaves zeros.  So each kernel computes
// interior + readable halos with an open/closed mask and the only
// communication left is the h/u/v ring exchange.
//
// Mapping: one cell per thread, blockIdx.x*256 covering columns (so a
// wave64's lanes touch consecutive columns -> fully coalesced), blockIdx.y
// covering rows; the benchmark grid launches ~27k workgroups, far above
// the 256-CU residency needed to fill all 8 XCDs.  Indices are 32-bit
// (int64 div/mod per element measured 2-4x off the HBM roofline).

#include <hip/hip_runtime.h>

#include "kernels.h"

namespace {

constexpr int kBlock = 256;

// XCD-aware block mapping: the dispatcher places block b on XCD b%8, so a
// plain (col-chunk, row) 2-D grid scatters adjacent rows across XCDs and
// every j±1 stencil read misses that XCD's L2.  The bijective remap below
// gives each XCD a contiguous band of row-major tiles, so neighbor rows
// are L2-resident (MI355X_MICROARCH.md §Workgroup dispatch; speed-only).
#define SW_BLOCK_MAP(ny, nx)                                               \
  const int gx_ = (nx + kBlock - 1) / kBlock;                              \
  const int T_ = (int)gridDim.x;                                           \
  const int t_ = (int)blockIdx.x;                                          \
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = t_ % 8, yc_ = t_ / 8;        \
  const int id_ =                                                          \
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)  \
      + yc_;                                                               \
  const int j = id_ / gx_;                                                 \
  const int i = (id_ % gx_) * kBlock + (int)threadIdx.x

constexpr float G = 9.81f;

struct SwFlags {
  // "open" = this halo receives neighbor data (exchange or periodic wrap)
  int south_open, north_open, west_open, east_open;
  // physical walls (reference shallow_water.py:258-262): u column nx-2
  // zeroed on a closed east edge, v row ny-2 zeroed on the north edge
  int east_wall, north_wall;
  // x halos are LOCAL periodic wraps (nproc_x == 1 && periodic), not
  // remote exchanges — lets the fused ring kernel synthesize them
  int x_wrap;
};

template <typename T>
struct SwArgs {
  T* fe;
  T* fn;
  T* q;
  T* ke;
  T* h;
  T* u;
  T* v;
  T* dnh;
  T* dnu;
  T* dnv;
  T* doh;
  T* dou;
  T* dov;
  T* h2;
  T* u2;
  T* v2;
  long long ny, nx;
  T dx, dy, dt, nu;
  T rdx, rdy;  // hoisted reciprocals (vector kernels: f32 division is
               // ~16-30 unpipelined cycles; uniform denominators multiply)
  T cor_base, cor_dj;  // coriolis(j) = cor_base + j * cor_dj
  T ab_a, ab_b;        // Adams-Bashforth coefficients (b=0 -> Euler)
  SwFlags f;
};

// hc = h padded by edge replication at closed halos (reference
// shallow_water.py:278-279 + enforce_boundaries); at open halos hc == h.
template <typename T>
__device__ inline T hc_at(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (j == 0 && !a.f.south_open) j = 1;
  if (j == ny - 1 && !a.f.north_open) j = ny - 2;
  if (i == 0 && !a.f.west_open) i = 1;
  if (i == nx - 1 && !a.f.east_open) i = nx - 2;
  return a.h[j * nx + i];
}

// stage 1: mass fluxes fe/fn, potential vorticity q, kinetic energy ke
template <typename T>
__global__ void sw_stage1_kernel(SwArgs<T> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  SW_BLOCK_MAP(ny, nx);
  if (i < nx && j < ny) {
    const int idx = j * nx + i;
    T u_ = a.u[idx], v_ = a.v[idx];
    T hcc = hc_at(a, j, i);

    // fe (u-kind): needs hc[j][i+1]
    T fe = T(0);
    if (i <= nx - 2 && (j >= 1 || a.f.south_open) &&
        (j <= ny - 2 || a.f.north_open) && (i >= 1 || a.f.west_open)) {
      fe = T(0.5) * (hcc + hc_at(a, j, i + 1)) * u_;
    }
    if (a.f.east_wall && i == nx - 2) fe = T(0);
    a.fe[idx] = fe;

    // fn (v-kind): needs hc[j+1][i]
    T fn = T(0);
    if (j <= ny - 2 && (i >= 1 || a.f.west_open) &&
        (i <= nx - 2 || a.f.east_open) && (j >= 1 || a.f.south_open)) {
      fn = T(0.5) * (hcc + hc_at(a, j + 1, i)) * v_;
    }
    if (a.f.north_wall && j == ny - 2) fn = T(0);
    a.fn[idx] = fn;

    // q (h-kind): needs v[j][i+1], u[j+1][i], hc[j..j+1][i..i+1]
    T q = T(0);
    if (i <= nx - 2 && j <= ny - 2 && (j >= 1 || a.f.south_open) &&
        (i >= 1 || a.f.west_open)) {
      T cor = a.cor_base + (T)j * a.cor_dj;
      q = cor + ((a.v[idx + 1] - v_) / a.dx - (a.u[idx + nx] - u_) / a.dy);
      q *= T(1) / (T(0.25) * (hcc + hc_at(a, j, i + 1) +
                              hc_at(a, j + 1, i) + hc_at(a, j + 1, i + 1)));
    }
    a.q[idx] = q;

    // ke (h-kind): needs u[j][i-1], v[j-1][i]
    T ke = T(0);
    if (i >= 1 && j >= 1 && (i <= nx - 2 || a.f.east_open) &&
        (j <= ny - 2 || a.f.north_open)) {
      T um = a.u[idx - 1], vm = a.v[idx - nx];
      ke = T(0.5) * (T(0.5) * (u_ * u_ + um * um) +
                     T(0.5) * (v_ * v_ + vm * vm));
    }
    a.ke[idx] = ke;
  }
}

// lateral-friction gradient helpers (masked per the eager eb() semantics)
template <typename T>
__device__ inline T gu_of_u(const SwArgs<T>& a, int j, int i) {
  // gu = nu * du/dx, u-kind halos: west col open-only, others unread
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (i > nx - 2 || (i == 0 && !a.f.west_open) || j < 1 || j > ny - 2)
    return T(0);
  if (a.f.east_wall && i == nx - 2) return T(0);
  return a.nu * (a.u[j * nx + i + 1] - a.u[j * nx + i]) / a.dx;
}

template <typename T>
__device__ inline T gv_of_u(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (j > ny - 2 || (j == 0 && !a.f.south_open) || i < 1 || i > nx - 2)
    return T(0);
  if (a.f.north_wall && j == ny - 2) return T(0);
  return a.nu * (a.u[(j + 1) * nx + i] - a.u[j * nx + i]) / a.dy;
}

template <typename T>
__device__ inline T gu_of_v(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (i > nx - 2 || (i == 0 && !a.f.west_open) || j < 1 || j > ny - 2)
    return T(0);
  if (a.f.east_wall && i == nx - 2) return T(0);
  return a.nu * (a.v[j * nx + i + 1] - a.v[j * nx + i]) / a.dx;
}

template <typename T>
__device__ inline T gv_of_v(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (j > ny - 2 || (j == 0 && !a.f.south_open) || i < 1 || i > nx - 2)
    return T(0);
  if (a.f.north_wall && j == ny - 2) return T(0);
  return a.nu * (a.v[(j + 1) * nx + i] - a.v[j * nx + i]) / a.dy;
}

// masked derived-field evaluators: single source of truth for the fe/fn/
// q/ke formulas AND their open/closed halo masks (same semantics as the
// stage-1 kernel, which remains as the two-pass fallback).  Evaluating
// these per use in the merged stage-8 kernel trades cheap cached loads +
// flops for a full write+read pass of four 26 MB arrays.
template <typename T>
__device__ inline T fe_at(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (i > nx - 2) return T(0);
  if (!((j >= 1 || a.f.south_open) && (j <= ny - 2 || a.f.north_open) &&
        (i >= 1 || a.f.west_open)))
    return T(0);
  if (a.f.east_wall && i == nx - 2) return T(0);
  return T(0.5) * (hc_at(a, j, i) + hc_at(a, j, i + 1)) * a.u[j * nx + i];
}

template <typename T>
__device__ inline T fn_at(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (j > ny - 2) return T(0);
  if (!((i >= 1 || a.f.west_open) && (i <= nx - 2 || a.f.east_open) &&
        (j >= 1 || a.f.south_open)))
    return T(0);
  if (a.f.north_wall && j == ny - 2) return T(0);
  return T(0.5) * (hc_at(a, j, i) + hc_at(a, j + 1, i)) * a.v[j * nx + i];
}

template <typename T>
__device__ inline T q_at(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (i > nx - 2 || j > ny - 2) return T(0);
  if (!((j >= 1 || a.f.south_open) && (i >= 1 || a.f.west_open)))
    return T(0);
  const int idx = j * nx + i;
  T cor = a.cor_base + (T)j * a.cor_dj;
  T q = cor + ((a.v[idx + 1] - a.v[idx]) / a.dx -
               (a.u[idx + nx] - a.u[idx]) / a.dy);
  q *= T(1) / (T(0.25) * (hc_at(a, j, i) + hc_at(a, j, i + 1) +
                          hc_at(a, j + 1, i) + hc_at(a, j + 1, i + 1)));
  return q;
}

template <typename T>
__device__ inline T ke_at(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  if (i < 1 || j < 1) return T(0);
  if (!((i <= nx - 2 || a.f.east_open) && (j <= ny - 2 || a.f.north_open)))
    return T(0);
  const int idx = j * nx + i;
  T u_ = a.u[idx], um = a.u[idx - 1];
  T v_ = a.v[idx], vm = a.v[idx - nx];
  return T(0.5) * (T(0.5) * (u_ * u_ + um * um) +
                   T(0.5) * (v_ * v_ + vm * vm));
}

// stage 8 = stage 1 + stage 6 fused: derived fields evaluated in-register,
// no fe/fn/q/ke array traffic at all.  The math lives in stage8_math so
// the fused update+friction stage (30) can evaluate a cell's update
// without storing it (same expressions, same order -> same rounding).
template <typename T>
struct CellUpd {
  T dnh, dnu, dnv, hh, uu, vv;
};

template <typename T>
__device__ inline CellUpd<T> stage8_math(const SwArgs<T>& a, int j, int i) {
  // caller guarantees an interior cell (1 <= j <= ny-2, 1 <= i <= nx-2)
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int idx = j * nx + i;
  T h_ = a.h[idx], u_ = a.u[idx], v_ = a.v[idx];
  CellUpd<T> r;

  T dnh = -(fe_at(a, j, i) - fe_at(a, j, i - 1)) / a.dx -
          (fn_at(a, j, i) - fn_at(a, j - 1, i)) / a.dy;
  r.dnh = dnh;

  T qc = q_at(a, j, i), qs = q_at(a, j - 1, i), qw = q_at(a, j, i - 1);
  T fnc = fn_at(a, j, i), fne = fn_at(a, j, i + 1);
  T fns = fn_at(a, j - 1, i), fnse = fn_at(a, j - 1, i + 1);
  T dnu = -G * (a.h[idx + 1] - h_) / a.dx +
          T(0.5) * (qc * T(0.5) * (fnc + fne) +
                    qs * T(0.5) * (fns + fnse));
  dnu -= (ke_at(a, j, i + 1) - ke_at(a, j, i)) / a.dx;
  r.dnu = dnu;

  T fec = fe_at(a, j, i), fen = fe_at(a, j + 1, i);
  T few = fe_at(a, j, i - 1), fenw = fe_at(a, j + 1, i - 1);
  T dnv = -G * (a.h[idx + nx] - h_) / a.dy -
          T(0.5) * (qc * T(0.5) * (fec + fen) +
                    qw * T(0.5) * (few + fenw));
  dnv -= (ke_at(a, j + 1, i) - ke_at(a, j, i)) / a.dy;
  r.dnv = dnv;

  T uu = u_ + a.dt * (a.ab_a * dnu + a.ab_b * a.dou[idx]);
  T vv = v_ + a.dt * (a.ab_a * dnv + a.ab_b * a.dov[idx]);
  r.hh = h_ + a.dt * (a.ab_a * dnh + a.ab_b * a.doh[idx]);
  if (a.f.east_wall && i == nx - 2) uu = T(0);
  if (a.f.north_wall && j == ny - 2) vv = T(0);
  r.uu = uu;
  r.vv = vv;
  return r;
}

template <typename T>
__device__ inline void stage8_cell(const SwArgs<T>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int idx = j * nx + i;
  if (j < 1 || j > ny - 2 || i < 1 || i > nx - 2) {
    a.h2[idx] = a.h[idx];
    a.u2[idx] = a.u[idx];
    a.v2[idx] = a.v[idx];
    return;
  }
  CellUpd<T> r = stage8_math(a, j, i);
  a.dnh[idx] = r.dnh;
  a.dnu[idx] = r.dnu;
  a.dnv[idx] = r.dnv;
  a.h2[idx] = r.hh;
  a.u2[idx] = r.uu;
  a.v2[idx] = r.vv;
}

template <typename T>
__global__ void sw_stage8_kernel(SwArgs<T> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  SW_BLOCK_MAP(ny, nx);
  if (i < nx && j < ny) {
    stage8_cell(a, j, i);
  }
}

// stage 6 = tendencies + AB time update merged with double-buffered field output:
// tendencies + AB/Euler update written to h2/u2/v2 — no in-place hazard,
// one full pass of traffic saved.
template <typename T>
__global__ void sw_stage6_kernel(SwArgs<T> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  SW_BLOCK_MAP(ny, nx);
  if (i < nx && j < ny) {
    const int idx = j * nx + i;
    T h_ = a.h[idx], u_ = a.u[idx], v_ = a.v[idx];
    if (j < 1 || j > ny - 2 || i < 1 || i > nx - 2) {
      // halo cells: fields pass through (exchange updates them next),
      // tendencies stay zero
      a.h2[idx] = h_;
      a.u2[idx] = u_;
      a.v2[idx] = v_;
      return;
    }

    T dnh = -(a.fe[idx] - a.fe[idx - 1]) / a.dx -
            (a.fn[idx] - a.fn[idx - nx]) / a.dy;
    a.dnh[idx] = dnh;

    T dnu = -G * (a.h[idx + 1] - h_) / a.dx +
            T(0.5) * (a.q[idx] * T(0.5) * (a.fn[idx] + a.fn[idx + 1]) +
                      a.q[idx - nx] * T(0.5) *
                          (a.fn[idx - nx] + a.fn[idx - nx + 1]));
    dnu -= (a.ke[idx + 1] - a.ke[idx]) / a.dx;
    a.dnu[idx] = dnu;

    T dnv = -G * (a.h[idx + nx] - h_) / a.dy -
            T(0.5) * (a.q[idx] * T(0.5) * (a.fe[idx] + a.fe[idx + nx]) +
                      a.q[idx - 1] * T(0.5) *
                          (a.fe[idx - 1] + a.fe[idx + nx - 1]));
    dnv -= (a.ke[idx + nx] - a.ke[idx]) / a.dy;
    a.dnv[idx] = dnv;

    T uu = u_ + a.dt * (a.ab_a * dnu + a.ab_b * a.dou[idx]);
    T vv = v_ + a.dt * (a.ab_a * dnv + a.ab_b * a.dov[idx]);
    a.h2[idx] = h_ + a.dt * (a.ab_a * dnh + a.ab_b * a.doh[idx]);
    if (a.f.east_wall && i == nx - 2) uu = T(0);
    if (a.f.north_wall && j == ny - 2) vv = T(0);
    a.u2[idx] = uu;
    a.v2[idx] = vv;
  }
}

// stage 7 = stage4 + stage5 merged: friction Laplacian computed inline
// (each g-gradient evaluated twice — FP is free, the array pass is not),
// updated u/v written to u2/v2.
template <typename T>
__global__ void sw_stage7_kernel(SwArgs<T> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  SW_BLOCK_MAP(ny, nx);
  if (i < nx && j < ny) {
    const int idx = j * nx + i;
    T u_ = a.u[idx], v_ = a.v[idx];
    if (j < 1 || j > ny - 2 || i < 1 || i > nx - 2) {
      a.u2[idx] = u_;
      a.v2[idx] = v_;
      return;
    }
    T lu = (gu_of_u(a, j, i) - gu_of_u(a, j, i - 1)) / a.dx +
           (gv_of_u(a, j, i) - gv_of_u(a, j - 1, i)) / a.dy;
    T lv = (gu_of_v(a, j, i) - gu_of_v(a, j, i - 1)) / a.dx +
           (gv_of_v(a, j, i) - gv_of_v(a, j - 1, i)) / a.dy;
    T uu = u_ + a.dt * lu;
    T vv = v_ + a.dt * lv;
    if (a.f.east_wall && i == nx - 2) uu = T(0);
    if (a.f.north_wall && j == ny - 2) vv = T(0);
    a.u2[idx] = uu;
    a.v2[idx] = vv;
  }
}


// ------------------------------------------------------- vectorized stages
// The scalar one-cell-per-thread stages measure ~3.5-4.5 TB/s: bound by
// load-instruction count (a ~20-load scalar stencil per cell), not HBM.
// These variants process 4 consecutive columns per thread with
// 4-byte-aligned float4 accesses (hipcc emits global_load_dwordx4 at dword
// alignment — verified on gfx950), cutting load instructions ~4x.  Edge
// packs (halo masks / walls / hc clamping) fall back to the scalar body,
// which stays the single source of truth for boundary semantics.

typedef float vf4 __attribute__((ext_vector_type(4), aligned(4)));

__device__ inline vf4 ld4(const float* p, long long off) {
  return *(const vf4*)(p + off);
}
__device__ inline void st4(float* p, long long off, vf4 v) {
  *(vf4*)(p + off) = v;
}

typedef float vf2 __attribute__((ext_vector_type(2), aligned(4)));

__device__ inline vf2 ld2(const float* p, long long off) {
  return *(const vf2*)(p + off);
}
__device__ inline void st2(float* p, long long off, vf2 v) {
  *(vf2*)(p + off) = v;
}
// float2 lane extracts from a float4 loaded at idx-1 (lane k = offset
// k-1): used by the 2-column stage variant
__device__ inline vf2 x01(vf4 A) { return (vf2){A.x, A.y}; }
__device__ inline vf2 x12(vf4 A) { return (vf2){A.y, A.z}; }
__device__ inline vf2 x23(vf4 A) { return (vf2){A.z, A.w}; }

// shifted-vector builders: lane c of the result holds value at i0+c+k
__device__ inline vf4 sh0(vf4 Am1, float x3) {
  // offset 0 from a load at idx-1 plus the i0+3 element
  return (vf4){Am1.y, Am1.z, Am1.w, x3};
}
__device__ inline vf4 sh1(vf4 Am1, float x3, float x4) {
  return (vf4){Am1.z, Am1.w, x3, x4};
}
__device__ inline vf4 sh0f(vf4 A0) { return A0; }  // load at idx
__device__ inline vf4 sh1f(vf4 A0, float x4) {
  return (vf4){A0.y, A0.z, A0.w, x4};
}
__device__ inline vf4 shm1f(float xm1, vf4 A0) {
  return (vf4){xm1, A0.x, A0.y, A0.z};
}

__device__ inline vf4 rcp4(vf4 x) {
  // v_rcp_f32 per lane (~1 ulp): removes correctly-rounded-division
  // chains (~30 unpipelined cycles each) from the per-pack critical path
  return (vf4){__builtin_amdgcn_rcpf(x.x), __builtin_amdgcn_rcpf(x.y),
               __builtin_amdgcn_rcpf(x.z), __builtin_amdgcn_rcpf(x.w)};
}
__device__ inline vf2 rcp2(vf2 x) {
  return (vf2){__builtin_amdgcn_rcpf(x.x), __builtin_amdgcn_rcpf(x.y)};
}


// scalar per-cell bodies (shared by scalar kernels' fallback)
__device__ inline void stage1_cell(const SwArgs<float>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int idx = j * nx + i;
  float u_ = a.u[idx], v_ = a.v[idx];
  float hcc = hc_at(a, j, i);
  float fe = 0.f;
  if (i <= nx - 2 && (j >= 1 || a.f.south_open) &&
      (j <= ny - 2 || a.f.north_open) && (i >= 1 || a.f.west_open)) {
    fe = 0.5f * (hcc + hc_at(a, j, i + 1)) * u_;
  }
  if (a.f.east_wall && i == nx - 2) fe = 0.f;
  a.fe[idx] = fe;
  float fn = 0.f;
  if (j <= ny - 2 && (i >= 1 || a.f.west_open) &&
      (i <= nx - 2 || a.f.east_open) && (j >= 1 || a.f.south_open)) {
    fn = 0.5f * (hcc + hc_at(a, j + 1, i)) * v_;
  }
  if (a.f.north_wall && j == ny - 2) fn = 0.f;
  a.fn[idx] = fn;
  float q = 0.f;
  if (i <= nx - 2 && j <= ny - 2 && (j >= 1 || a.f.south_open) &&
      (i >= 1 || a.f.west_open)) {
    float cor = a.cor_base + (float)j * a.cor_dj;
    q = cor + ((a.v[idx + 1] - v_) / a.dx - (a.u[idx + nx] - u_) / a.dy);
    q *= 1.f / (0.25f * (hcc + hc_at(a, j, i + 1) + hc_at(a, j + 1, i) +
                         hc_at(a, j + 1, i + 1)));
  }
  a.q[idx] = q;
  float ke = 0.f;
  if (i >= 1 && j >= 1 && (i <= nx - 2 || a.f.east_open) &&
      (j <= ny - 2 || a.f.north_open)) {
    float um = a.u[idx - 1], vm = a.v[idx - nx];
    ke = 0.5f * (0.5f * (u_ * u_ + um * um) + 0.5f * (v_ * v_ + vm * vm));
  }
  a.ke[idx] = ke;
}

__device__ inline void stage6_cell(const SwArgs<float>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int idx = j * nx + i;
  float h_ = a.h[idx], u_ = a.u[idx], v_ = a.v[idx];
  if (j < 1 || j > ny - 2 || i < 1 || i > nx - 2) {
    a.h2[idx] = h_;
    a.u2[idx] = u_;
    a.v2[idx] = v_;
    return;
  }
  float dnh = -(a.fe[idx] - a.fe[idx - 1]) / a.dx -
              (a.fn[idx] - a.fn[idx - nx]) / a.dy;
  a.dnh[idx] = dnh;
  float dnu = -G * (a.h[idx + 1] - h_) / a.dx +
              0.5f * (a.q[idx] * 0.5f * (a.fn[idx] + a.fn[idx + 1]) +
                      a.q[idx - nx] * 0.5f *
                          (a.fn[idx - nx] + a.fn[idx - nx + 1]));
  dnu -= (a.ke[idx + 1] - a.ke[idx]) / a.dx;
  a.dnu[idx] = dnu;
  float dnv = -G * (a.h[idx + nx] - h_) / a.dy -
              0.5f * (a.q[idx] * 0.5f * (a.fe[idx] + a.fe[idx + nx]) +
                      a.q[idx - 1] * 0.5f *
                          (a.fe[idx - 1] + a.fe[idx + nx - 1]));
  dnv -= (a.ke[idx + nx] - a.ke[idx]) / a.dy;
  a.dnv[idx] = dnv;
  float uu = u_ + a.dt * (a.ab_a * dnu + a.ab_b * a.dou[idx]);
  float vv = v_ + a.dt * (a.ab_a * dnv + a.ab_b * a.dov[idx]);
  a.h2[idx] = h_ + a.dt * (a.ab_a * dnh + a.ab_b * a.doh[idx]);
  if (a.f.east_wall && i == nx - 2) uu = 0.f;
  if (a.f.north_wall && j == ny - 2) vv = 0.f;
  a.u2[idx] = uu;
  a.v2[idx] = vv;
}

__device__ inline void stage7_cell(const SwArgs<float>& a, int j, int i) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int idx = j * nx + i;
  float u_ = a.u[idx], v_ = a.v[idx];
  if (j < 1 || j > ny - 2 || i < 1 || i > nx - 2) {
    a.u2[idx] = u_;
    a.v2[idx] = v_;
    return;
  }
  float lu = (gu_of_u(a, j, i) - gu_of_u(a, j, i - 1)) / a.dx +
             (gv_of_u(a, j, i) - gv_of_u(a, j - 1, i)) / a.dy;
  float lv = (gu_of_v(a, j, i) - gu_of_v(a, j, i - 1)) / a.dx +
             (gv_of_v(a, j, i) - gv_of_v(a, j - 1, i)) / a.dy;
  float uu = u_ + a.dt * lu;
  float vv = v_ + a.dt * lv;
  if (a.f.east_wall && i == nx - 2) uu = 0.f;
  if (a.f.north_wall && j == ny - 2) vv = 0.f;
  a.u2[idx] = uu;
  a.v2[idx] = vv;
}

// simpler pack mapping: thread t covers row j = t / packs_per_row,
// columns [4*(t % ppr), 4*(t % ppr)+3]
__global__ void sw_stage1v(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 3) / 4;
  // XCD-aware block remap (same rationale as SW_BLOCK_MAP): give each XCD
  // a contiguous band of rows so j±1 stencil rows hit that XCD's L2
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int t = bid_ * (int)blockDim.x + (int)threadIdx.x;
  if (t >= ppr * ny) return;
  const int j = t / ppr;
  const int i0 = (t % ppr) * 4;
  const int jmax = (a.f.north_open && !a.f.north_wall) ? ny - 2 : ny - 3;
  const int imax = (a.f.east_open && !a.f.east_wall) ? nx - 1 : nx - 2;
  const int imin = a.f.west_open ? 0 + 1 : 2;  // cols use i-1 only in ke
  const bool fast = j >= 1 && j <= jmax && i0 >= imin && i0 + 4 <= imax &&
                    a.f.west_open && a.f.east_open;
  if (!fast) {
    for (int c = 0; c < 4 && i0 + c < nx; ++c) stage1_cell(a, j, i0 + c);
    return;
  }
  const long long idx = (long long)j * nx + i0;
  vf4 hA = ld4(a.h, idx), hAe = ld4(a.h, idx + 1);
  vf4 hB = ld4(a.h, idx + nx), hBe = ld4(a.h, idx + nx + 1);
  vf4 uc = ld4(a.u, idx), uw = ld4(a.u, idx - 1), un = ld4(a.u, idx + nx);
  vf4 vc = ld4(a.v, idx), ve = ld4(a.v, idx + 1), vs = ld4(a.v, idx - nx);
  st4(a.fe, idx, 0.5f * (hA + hAe) * uc);
  st4(a.fn, idx, 0.5f * (hA + hB) * vc);
  float cor = a.cor_base + (float)j * a.cor_dj;
  vf4 q = cor + ((ve - vc) * a.rdx - (un - uc) * a.rdy);
  q *= rcp4(0.25f * (hA + hAe + hB + hBe));
  st4(a.q, idx, q);
  vf4 ke = 0.5f * (0.5f * (uc * uc + uw * uw) + 0.5f * (vc * vc + vs * vs));
  st4(a.ke, idx, ke);
}

// vector stage 6 (tendencies + AB update into the alt buffers)
__global__ void sw_stage6v(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 3) / 4;
  // XCD-aware block remap (same rationale as SW_BLOCK_MAP): give each XCD
  // a contiguous band of rows so j±1 stencil rows hit that XCD's L2
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int t = bid_ * (int)blockDim.x + (int)threadIdx.x;
  if (t >= ppr * ny) return;
  const int j = t / ppr;
  const int i0 = (t % ppr) * 4;
  const int jmax = a.f.north_wall ? ny - 3 : ny - 2;
  const int imax = a.f.east_wall ? nx - 2 : nx - 1;
  const bool fast = j >= 1 && j <= jmax && i0 >= 1 && i0 + 4 <= imax;
  if (!fast) {
    for (int c = 0; c < 4 && i0 + c < nx; ++c) stage6_cell(a, j, i0 + c);
    return;
  }
  const long long idx = (long long)j * nx + i0;
  vf4 fec = ld4(a.fe, idx), few = ld4(a.fe, idx - 1);
  vf4 fen = ld4(a.fe, idx + nx), fenw = ld4(a.fe, idx + nx - 1);
  vf4 fnc = ld4(a.fn, idx), fne = ld4(a.fn, idx + 1);
  vf4 fns = ld4(a.fn, idx - nx), fnse = ld4(a.fn, idx - nx + 1);
  vf4 qc = ld4(a.q, idx), qs = ld4(a.q, idx - nx), qw = ld4(a.q, idx - 1);
  vf4 kec = ld4(a.ke, idx), kee = ld4(a.ke, idx + 1),
      ken = ld4(a.ke, idx + nx);
  vf4 hc = ld4(a.h, idx), he = ld4(a.h, idx + 1), hn = ld4(a.h, idx + nx);
  vf4 uc = ld4(a.u, idx), vc = ld4(a.v, idx);
  vf4 doh = ld4(a.doh, idx), dou = ld4(a.dou, idx), dov = ld4(a.dov, idx);

  vf4 dnh = -(fec - few) * a.rdx - (fnc - fns) * a.rdy;
  st4(a.dnh, idx, dnh);
  vf4 dnu = -G * (he - hc) * a.rdx +
            0.5f * (qc * 0.5f * (fnc + fne) + qs * 0.5f * (fns + fnse));
  dnu -= (kee - kec) * a.rdx;
  st4(a.dnu, idx, dnu);
  vf4 dnv = -G * (hn - hc) * a.rdy -
            0.5f * (qc * 0.5f * (fec + fen) + qw * 0.5f * (few + fenw));
  dnv -= (ken - kec) * a.rdy;
  st4(a.dnv, idx, dnv);

  st4(a.h2, idx, hc + a.dt * (a.ab_a * dnh + a.ab_b * doh));
  st4(a.u2, idx, uc + a.dt * (a.ab_a * dnu + a.ab_b * dou));
  st4(a.v2, idx, vc + a.dt * (a.ab_a * dnv + a.ab_b * dov));
}

// vector stage 7 (friction)
__global__ void sw_stage7v(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 3) / 4;
  // XCD-aware block remap (same rationale as SW_BLOCK_MAP): give each XCD
  // a contiguous band of rows so j±1 stencil rows hit that XCD's L2
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int t = bid_ * (int)blockDim.x + (int)threadIdx.x;
  if (t >= ppr * ny) return;
  const int j = t / ppr;
  const int i0 = (t % ppr) * 4;
  const int jmin = a.f.south_open ? 1 : 2;
  const int jmax = (a.f.north_open && !a.f.north_wall) ? ny - 2 : ny - 3;
  const int imin = a.f.west_open ? 1 : 2;
  const int imax = (a.f.east_open && !a.f.east_wall) ? nx - 1 : nx - 2;
  const bool fast = j >= jmin && j <= jmax && i0 >= imin &&
                    i0 + 4 <= imax && i0 + 5 <= nx;
  if (!fast) {
    for (int c = 0; c < 4 && i0 + c < nx; ++c) stage7_cell(a, j, i0 + c);
    return;
  }
  const long long idx = (long long)j * nx + i0;
  // 8 loads instead of 10: center/west/east derived from one unaligned
  // ld4 + ld2 pair per row (same trick as stage18v)
  vf4 uA = ld4(a.u, idx - 1);
  vf2 uB = ld2(a.u, idx + 3);
  vf4 un = ld4(a.u, idx + nx), us = ld4(a.u, idx - nx);
  vf4 vA = ld4(a.v, idx - 1);
  vf2 vB = ld2(a.v, idx + 3);
  vf4 vn = ld4(a.v, idx + nx), vs = ld4(a.v, idx - nx);
  vf4 uw = uA, uc = sh0(uA, uB.x), ue = sh1(uA, uB.x, uB.y);
  vf4 vw = vA, vcc = sh0(vA, vB.x), ve = sh1(vA, vB.x, vB.y);
  const float nu = a.nu;
  vf4 lu = (nu * (ue - uc) * a.rdx - nu * (uc - uw) * a.rdx) * a.rdx +
           (nu * (un - uc) * a.rdy - nu * (uc - us) * a.rdy) * a.rdy;
  vf4 lv = (nu * (ve - vcc) * a.rdx - nu * (vcc - vw) * a.rdx) * a.rdx +
           (nu * (vn - vcc) * a.rdy - nu * (vcc - vs) * a.rdy) * a.rdy;
  st4(a.u2, idx, uc + a.dt * lu);
  st4(a.v2, idx, vcc + a.dt * lv);
}


// vector stage 18 = merged stage 8 vectorized: tendencies + update
// computed straight from h/u/v with float4 packs and shifted vectors —
// 312 MB/step of HBM traffic instead of the two-pass 598 MB.
__global__ void sw_stage18v(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 3) / 4;
  // XCD-aware block remap (same rationale as SW_BLOCK_MAP): give each XCD
  // a contiguous band of rows so j±1 stencil rows hit that XCD's L2
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int t = bid_ * (int)blockDim.x + (int)threadIdx.x;
  if (t >= ppr * ny) return;
  const int j = t / ppr;
  const int i0 = (t % ppr) * 4;
  const int jmin = a.f.south_open ? 1 : 2;
  const int jmax = (a.f.north_open && !a.f.north_wall) ? ny - 2 : ny - 3;
  const int imin = a.f.west_open ? 1 : 2;
  const int imax = (a.f.east_open && !a.f.east_wall) ? nx - 1 : nx - 2;
  const bool fast = j >= jmin && j <= jmax && i0 >= imin && i0 + 4 <= imax;
  if (!fast) {
    for (int c = 0; c < 4 && i0 + c < nx; ++c) stage8_cell(a, j, i0 + c);
    return;
  }
  const long long idx = (long long)j * nx + i0;
  const long long idn = idx - nx, idp = idx + nx;

  // issue ALL global loads first (one deep in-flight window, a single
  // vmcnt wait region), then build the shifted vectors
  vf4 HmA = ld4(a.h, idn);
  float Hm4 = a.h[idn + 4];
  vf4 H0A = ld4(a.h, idx - 1);
  vf2 H0B = ld2(a.h, idx + 3);
  vf4 HpA = ld4(a.h, idp - 1);
  vf2 HpB = ld2(a.h, idp + 3);
  vf4 Um0 = ld4(a.u, idn);
  vf4 U0A = ld4(a.u, idx - 1);
  vf2 U0B = ld2(a.u, idx + 3);
  vf4 UpA = ld4(a.u, idp);
  float Upm1s = a.u[idp - 1];
  vf4 VmA = ld4(a.v, idn);
  float Vm4 = a.v[idn + 4];
  vf4 V0A = ld4(a.v, idx - 1);
  vf2 V0B = ld2(a.v, idx + 3);
  vf4 Vp0 = ld4(a.v, idp);
  vf4 doh = ld4(a.doh, idx), dou = ld4(a.dou, idx), dov = ld4(a.dov, idx);

  // shifted-vector views: h rows j-1 (offsets 0,1), j / j+1 (-1,0,1);
  // u rows j-1 (0), j (-1,0,1), j+1 (-1,0); v rows j-1 (0,1), j (-1,0,1),
  // j+1 (0)
  vf4 Hm0 = sh0f(HmA), Hm1 = sh1f(HmA, Hm4);
  vf4 H0m1 = H0A, H00 = sh0(H0A, H0B.x), H01 = sh1(H0A, H0B.x, H0B.y);
  vf4 Hpm1 = HpA, Hp0 = sh0(HpA, HpB.x), Hp1 = sh1(HpA, HpB.x, HpB.y);
  vf4 U0m1 = U0A, U00 = sh0(U0A, U0B.x), U01 = sh1(U0A, U0B.x, U0B.y);
  vf4 Up0 = UpA, Upm1 = shm1f(Upm1s, UpA);
  vf4 Vm0 = VmA, Vm1 = sh1f(VmA, Vm4);
  vf4 V0m1 = V0A, V00 = sh0(V0A, V0B.x), V01 = sh1(V0A, V0B.x, V0B.y);

  const float rdx = a.rdx, rdy = a.rdy;

  // derived fields (same formulas as fe_at/fn_at/q_at/ke_at, full-mask
  // region so every value is the plain formula)
  vf4 fe_c = 0.5f * (H00 + H01) * U00;
  vf4 fe_w = 0.5f * (H0m1 + H00) * U0m1;
  vf4 fe_n = 0.5f * (Hp0 + Hp1) * Up0;
  vf4 fe_nw = 0.5f * (Hpm1 + Hp0) * Upm1;
  vf4 fn_c = 0.5f * (H00 + Hp0) * V00;
  vf4 fn_e = 0.5f * (H01 + Hp1) * V01;
  vf4 fn_s = 0.5f * (Hm0 + H00) * Vm0;
  vf4 fn_se = 0.5f * (Hm1 + H01) * Vm1;

  float corj = a.cor_base + (float)j * a.cor_dj;
  float corjm = a.cor_base + (float)(j - 1) * a.cor_dj;
  vf4 q_c = corj + ((V01 - V00) * rdx - (Up0 - U00) * rdy);
  q_c *= rcp4(0.25f * (H00 + H01 + Hp0 + Hp1));
  vf4 q_s = corjm + ((Vm1 - Vm0) * rdx - (U00 - Um0) * rdy);
  q_s *= rcp4(0.25f * (Hm0 + Hm1 + H00 + H01));
  vf4 q_w = corj + ((V00 - V0m1) * rdx - (Upm1 - U0m1) * rdy);
  q_w *= rcp4(0.25f * (H0m1 + H00 + Hpm1 + Hp0));

  vf4 ke_c = 0.5f * (0.5f * (U00 * U00 + U0m1 * U0m1) +
                     0.5f * (V00 * V00 + Vm0 * Vm0));
  vf4 ke_e = 0.5f * (0.5f * (U01 * U01 + U00 * U00) +
                     0.5f * (V01 * V01 + Vm1 * Vm1));
  vf4 ke_n = 0.5f * (0.5f * (Up0 * Up0 + Upm1 * Upm1) +
                     0.5f * (Vp0 * Vp0 + V00 * V00));

  vf4 dnh = -(fe_c - fe_w) * rdx - (fn_c - fn_s) * rdy;
  vf4 dnu = -G * (H01 - H00) * rdx +
            0.5f * (q_c * 0.5f * (fn_c + fn_e) +
                    q_s * 0.5f * (fn_s + fn_se));
  dnu -= (ke_e - ke_c) * rdx;
  vf4 dnv = -G * (Hp0 - H00) * rdy -
            0.5f * (q_c * 0.5f * (fe_c + fe_n) +
                    q_w * 0.5f * (fe_w + fe_nw));
  dnv -= (ke_n - ke_c) * rdy;

  st4(a.dnh, idx, dnh);
  st4(a.dnu, idx, dnu);
  st4(a.dnv, idx, dnv);
  st4(a.h2, idx, H00 + a.dt * (a.ab_a * dnh + a.ab_b * doh));
  st4(a.u2, idx, U00 + a.dt * (a.ab_a * dnu + a.ab_b * dou));
  st4(a.v2, idx, V00 + a.dt * (a.ab_a * dnv + a.ab_b * dov));
}

// stage 27 = stage 17 (friction Laplacian) at 2 columns/thread — same
// occupancy rationale as stage 19 below.
__device__ inline void stage27_pair(const SwArgs<float>& a, int j,
                                    int i0) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int jmin = a.f.south_open ? 1 : 2;
  const int jmax = (a.f.north_open && !a.f.north_wall) ? ny - 2 : ny - 3;
  const int imin = a.f.west_open ? 1 : 2;
  const int imax = (a.f.east_open && !a.f.east_wall) ? nx - 1 : nx - 2;
  const bool fast = j >= jmin && j <= jmax && i0 >= imin &&
                    i0 + 2 <= imax && i0 + 2 < nx;
  if (!fast) {
    for (int c = 0; c < 2 && i0 + c < nx; ++c) stage7_cell(a, j, i0 + c);
    return;
  }
  const long long idx = (long long)j * nx + i0;
  vf4 U0A = ld4(a.u, idx - 1);
  vf2 un = ld2(a.u, idx + nx), us = ld2(a.u, idx - nx);
  vf4 V0A = ld4(a.v, idx - 1);
  vf2 vn = ld2(a.v, idx + nx), vs = ld2(a.v, idx - nx);
  vf2 uw = x01(U0A), uc = x12(U0A), ue = x23(U0A);
  vf2 vw = x01(V0A), vcc = x12(V0A), ve = x23(V0A);
  const float nu = a.nu;
  vf2 lu = (nu * (ue - uc) * a.rdx - nu * (uc - uw) * a.rdx) * a.rdx +
           (nu * (un - uc) * a.rdy - nu * (uc - us) * a.rdy) * a.rdy;
  vf2 lv = (nu * (ve - vcc) * a.rdx - nu * (vcc - vw) * a.rdx) * a.rdx +
           (nu * (vn - vcc) * a.rdy - nu * (vcc - vs) * a.rdy) * a.rdy;
  st2(a.u2, idx, uc + a.dt * lu);
  st2(a.v2, idx, vcc + a.dt * lv);
}

__global__ void sw_stage27v(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 1) / 2;
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int t = bid_ * (int)blockDim.x + (int)threadIdx.x;
  if (t >= ppr * ny) return;
  stage27_pair(a, t / ppr, (t % ppr) * 2);
}

// halo/compute overlap split of the friction stage: a pair (2 cells) is
// halo-INDEPENDENT iff its 3x4 input stencil stays inside the interior
// ring [1, ny-2] x [1, nx-2] - those pairs (stage 28) can run while the
// halo exchange is still in flight on a second stream; the remaining
// ring pairs (stage 29) run after the exchange joins.  Both call the
// identical stage27_pair, so 28+29 together reproduce stage 27 BITWISE.
__device__ inline bool stage27_pair_independent(const SwArgs<float>& a,
                                                int j, int i0) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  return j >= 2 && j <= ny - 3 && i0 >= 2 && i0 + 2 <= nx - 2;
}

__global__ void sw_stage28v(SwArgs<float> a) {  // interior pairs only
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 1) / 2;
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int t = bid_ * (int)blockDim.x + (int)threadIdx.x;
  if (t >= ppr * ny) return;
  const int j = t / ppr, i0 = (t % ppr) * 2;
  if (!stage27_pair_independent(a, j, i0)) return;
  stage27_pair(a, j, i0);
}

__global__ void sw_stage29v(SwArgs<float> a) {  // ring pairs only
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 1) / 2;
  // ring pairs: rows 0,1,ny-3..ny-1 full width + 2 pair-columns on each
  // side of the middle rows - enumerated compactly so the launch is tiny
  const int side_pairs = 2;  // pairs at i0 = 0,2 and the last two pairs
  const int mid_rows = ny - 5 > 0 ? ny - 5 : 0;
  const long long top = 5LL * ppr;  // rows 0,1 and ny-3,ny-2,ny-1
  const long long ring = top + (long long)mid_rows * 2 * side_pairs;
  const long long t =
      (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= ring) return;
  int j, i0;
  if (t < top) {
    const int r = (int)(t / ppr);       // 0,1 -> rows 0,1; 2,3,4 -> top
    j = r < 2 ? r : ny - 5 + r;         // rows ny-3, ny-2, ny-1
    i0 = (int)(t % ppr) * 2;
  } else {
    const long long u = t - top;
    const int row = (int)(u / (2 * side_pairs));
    const int k = (int)(u % (2 * side_pairs));
    j = 2 + row;
    const int last0 = ((nx + 1) / 2 - 1) * 2;  // final pair start
    i0 = k < side_pairs ? k * 2 : last0 - (k - side_pairs) * 2;
  }
  if (stage27_pair_independent(a, j, i0)) return;  // guard small domains
  stage27_pair(a, j, i0);
}

// stage 19 = stage 18 at 2 columns/thread (float2 math, float4 row
// loads).  stage18v holds 118 VGPRs -> 4 waves/SIMD and parks ~48% of
// cycles on memory (profiles/README.md); halving the per-thread state
// trades ILP for occupancy to hide that latency.  Same operation order
// as stage18v, so the trajectories agree to float rounding.
//
// NT variant (stage 20): the do* tendency loads and every store are
// single-use streams consumed only on the NEXT step (by which point the
// whole 312 MB working set has cycled through anyway), so nontemporal
// hints keep them from evicting the h/u/v rows that DO have intra-step
// reuse across j-neighboring threads.
template <bool NT>
__device__ inline vf2 ld2s(const float* p, long long off) {
  if constexpr (NT) {
    return __builtin_nontemporal_load((const vf2*)(p + off));
  } else {
    return *(const vf2*)(p + off);
  }
}
template <bool NT>
__device__ inline void st2s(float* p, long long off, vf2 v) {
  if constexpr (NT) {
    __builtin_nontemporal_store(v, (vf2*)(p + off));
  } else {
    *(vf2*)(p + off) = v;
  }
}

// the pair-fast predicate of the 2-column update kernels: matches
// stage19_cells' fast branch exactly (the ld4 at idn reads offsets 0..3,
// so the pair must keep i0+3 in-row)
__device__ inline bool stage19_pair_fast(const SwArgs<float>& a, int j,
                                         int i0) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int jmin = a.f.south_open ? 1 : 2;
  const int jmax = (a.f.north_open && !a.f.north_wall) ? ny - 2 : ny - 3;
  const int imin = a.f.west_open ? 1 : 2;
  const int imax = (a.f.east_open && !a.f.east_wall) ? nx - 1 : nx - 2;
  return j >= jmin && j <= jmax && i0 >= imin && i0 + 2 <= imax &&
         i0 + 3 < nx;
}

struct Upd2 {
  vf2 dnh, dnu, dnv, hh, uu, vv;
};

// fast-region update math for one 2-column pack (the stage19 body);
// caller guarantees stage19_pair_fast(a, j, i0)
template <bool NT>
__device__ inline Upd2 stage19_math(const SwArgs<float>& a, int j,
                                    int i0) {
  const int nx = (int)a.nx;
  const long long idx = (long long)j * nx + i0;
  const long long idn = idx - nx, idp = idx + nx;

  // all loads first (one vmcnt window), float4 per row
  vf4 HmA = ld4(a.h, idn);      // h[j-1][0..3]
  vf4 H0A = ld4(a.h, idx - 1);  // h[j][-1..2]
  vf4 HpA = ld4(a.h, idp - 1);  // h[j+1][-1..2]
  vf2 Um0 = ld2(a.u, idn);      // u[j-1][0..1]
  vf4 U0A = ld4(a.u, idx - 1);  // u[j][-1..2]
  vf4 UpA = ld4(a.u, idp - 1);  // u[j+1][-1..2]
  vf4 VmA = ld4(a.v, idn);      // v[j-1][0..3]
  vf4 V0A = ld4(a.v, idx - 1);  // v[j][-1..2]
  vf2 Vp0 = ld2(a.v, idp);      // v[j+1][0..1]
  vf2 doh = ld2s<NT>(a.doh, idx), dou = ld2s<NT>(a.dou, idx),
      dov = ld2s<NT>(a.dov, idx);

  vf2 Hm0 = x01(HmA), Hm1 = x12(HmA);
  vf2 H0m1 = x01(H0A), H00 = x12(H0A), H01 = x23(H0A);
  vf2 Hpm1 = x01(HpA), Hp0 = x12(HpA), Hp1 = x23(HpA);
  vf2 U0m1 = x01(U0A), U00 = x12(U0A), U01 = x23(U0A);
  vf2 Upm1 = x01(UpA), Up0 = x12(UpA);
  vf2 Vm0 = x01(VmA), Vm1 = x12(VmA);
  vf2 V0m1 = x01(V0A), V00 = x12(V0A), V01 = x23(V0A);

  const float rdx = a.rdx, rdy = a.rdy;

  vf2 fe_c = 0.5f * (H00 + H01) * U00;
  vf2 fe_w = 0.5f * (H0m1 + H00) * U0m1;
  vf2 fe_n = 0.5f * (Hp0 + Hp1) * Up0;
  vf2 fe_nw = 0.5f * (Hpm1 + Hp0) * Upm1;
  vf2 fn_c = 0.5f * (H00 + Hp0) * V00;
  vf2 fn_e = 0.5f * (H01 + Hp1) * V01;
  vf2 fn_s = 0.5f * (Hm0 + H00) * Vm0;
  vf2 fn_se = 0.5f * (Hm1 + H01) * Vm1;

  float corj = a.cor_base + (float)j * a.cor_dj;
  float corjm = a.cor_base + (float)(j - 1) * a.cor_dj;
  vf2 q_c = corj + ((V01 - V00) * rdx - (Up0 - U00) * rdy);
  q_c *= rcp2(0.25f * (H00 + H01 + Hp0 + Hp1));
  vf2 q_s = corjm + ((Vm1 - Vm0) * rdx - (U00 - Um0) * rdy);
  q_s *= rcp2(0.25f * (Hm0 + Hm1 + H00 + H01));
  vf2 q_w = corj + ((V00 - V0m1) * rdx - (Upm1 - U0m1) * rdy);
  q_w *= rcp2(0.25f * (H0m1 + H00 + Hpm1 + Hp0));

  vf2 ke_c = 0.5f * (0.5f * (U00 * U00 + U0m1 * U0m1) +
                     0.5f * (V00 * V00 + Vm0 * Vm0));
  vf2 ke_e = 0.5f * (0.5f * (U01 * U01 + U00 * U00) +
                     0.5f * (V01 * V01 + Vm1 * Vm1));
  vf2 ke_n = 0.5f * (0.5f * (Up0 * Up0 + Upm1 * Upm1) +
                     0.5f * (Vp0 * Vp0 + V00 * V00));

  vf2 dnh = -(fe_c - fe_w) * rdx - (fn_c - fn_s) * rdy;
  vf2 dnu = -G * (H01 - H00) * rdx +
            0.5f * (q_c * 0.5f * (fn_c + fn_e) +
                    q_s * 0.5f * (fn_s + fn_se));
  dnu -= (ke_e - ke_c) * rdx;
  vf2 dnv = -G * (Hp0 - H00) * rdy -
            0.5f * (q_c * 0.5f * (fe_c + fe_n) +
                    q_w * 0.5f * (fe_w + fe_nw));
  dnv -= (ke_n - ke_c) * rdy;

  Upd2 r;
  r.dnh = dnh;
  r.dnu = dnu;
  r.dnv = dnv;
  r.hh = H00 + a.dt * (a.ab_a * dnh + a.ab_b * doh);
  r.uu = U00 + a.dt * (a.ab_a * dnu + a.ab_b * dou);
  r.vv = V00 + a.dt * (a.ab_a * dnv + a.ab_b * dov);
  return r;
}

template <bool NT>
__device__ inline void stage19_cells(const SwArgs<float>& a, int j,
                                     int i0) {
  const int nx = (int)a.nx;
  if (!stage19_pair_fast(a, j, i0)) {
    for (int c = 0; c < 2 && i0 + c < nx; ++c) stage8_cell(a, j, i0 + c);
    return;
  }
  const long long idx = (long long)j * nx + i0;
  Upd2 r = stage19_math<NT>(a, j, i0);
  st2s<NT>(a.dnh, idx, r.dnh);
  st2s<NT>(a.dnu, idx, r.dnu);
  st2s<NT>(a.dnv, idx, r.dnv);
  st2s<NT>(a.h2, idx, r.hh);
  st2s<NT>(a.u2, idx, r.uu);
  st2s<NT>(a.v2, idx, r.vv);
}

template <bool NT>
__global__ void sw_stage19t(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 1) / 2;
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int t = bid_ * (int)blockDim.x + (int)threadIdx.x;
  if (t >= ppr * ny) return;
  stage19_cells<NT>(a, t / ppr, (t % ppr) * 2);
}

// stage 21/22/23 = stage 19 with a 2-D band-tile block mapping (TJ = 4,
// 8, 16 rows per 256-thread block): each block covers TJ consecutive
// rows x (512/TJ) columns, so the j+-1 stencil rows are re-read from the
// CU's own L1/L2 slice instead of crossing to the XCD L2.  The kernel is
// already at minimum HBM traffic (profiles/README.md) and latency-bound;
// tiling attacks average load latency.  Tile ids stay row-band ordered
// under the XCD remap so each XCD still owns a contiguous band.
template <int TJ>
__global__ void sw_stage21t(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int ppr = (nx + 1) / 2;
  constexpr int TPC = 256 / TJ;  // thread-pairs (2 cols each) per tile row
  const int nti = (ppr + TPC - 1) / TPC;
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int tj = bid_ / nti, ti = bid_ % nti;
  const int jj = (int)threadIdx.x / TPC, ii = (int)threadIdx.x % TPC;
  const int j = tj * TJ + jj;
  const int i0 = (ti * TPC + ii) * 2;
  if (j >= ny || i0 >= nx) return;
  stage19_cells<false>(a, j, i0);
}

// ------------------------------------------------ stage 30: fused step
// Update + friction in one pass (world-1 / fully-local halos only).
// The two-kernel pipeline writes the post-update u'/v' to HBM, wrap-
// exchanges their halos, and reads them back for the friction Laplacian:
// 4 of the step's 16 field passes plus one whole exchange exist only to
// ferry that intermediate.  Stage 30 keeps u'/v' in LDS instead:
// blocks compute the stage-19 update for a (TJ+2)-row tile into LDS
// (vertical recompute replaces the cross-block dependency) and apply
// the friction to the TJ middle rows from LDS.
//
// Numerics: same formulas in the same order as the two-kernel path
// (stage19_math / stage8_math / stage27's rdx form), with local wrap
// halos synthesized via s30_read_uv.  NOT bitwise: the compiler
// contracts the shared expression trees to FMA differently per inlining
// site (~1 ulp/step, tolerance-tested in tests/test_gpu_ops.py::
// test_stage30_matches_two_kernel_path).

// The boundary ring's friction needs the post-update u'/v' — instead of
// recomputing them per stencil point (measured 26 us/step), the fast
// kernel and the update-ring kernel STORE the ring strip's u'/v' into
// the fe/fn scratch fields (unused on the fused path), and the
// friction-ring kernel just reads them.  Ring strip = the cleanup
// enumeration: rows 0..3 / ny-4..ny-1, cols 0..5 / nx-6..nx-1 — covers
// every not-s30_fast_out cell, its 4 stencil neighbors, and the wrap
// sources (rows/cols 1, ny-2, nx-2).
__device__ inline bool s30_in_ring(const SwArgs<float>& a, int jj,
                                   int i0) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  return jj <= 3 || jj >= ny - 4 || i0 <= 5 || i0 >= nx - 7;
}

// what the ring friction reads at (jj, ii): the mid-step exchange's
// value.  Local periodic x halos (x_wrap) map to their wrap source;
// remote open halos read the EXCHANGED fe/fn strip (the N>1 fused step
// runs a real fe/fn halo exchange between ringA and ringB);
// closed-edge halos pass the pre-update field through; interior cells
// read the staged strip.
__device__ inline void s30_read_uv(const SwArgs<float>& a, int jj, int ii,
                                   float* uu, float* vv) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  int is = ii;
  if (a.f.x_wrap) {
    if (ii == 0) is = nx - 2;
    else if (ii == nx - 1) is = 1;
  }
  const long long idx = (long long)jj * nx + is;
  const bool closed = (jj == 0 && !a.f.south_open) ||
                      (jj == ny - 1 && !a.f.north_open) ||
                      (is == 0 && !a.f.west_open) ||
                      (is == nx - 1 && !a.f.east_open);
  if (closed) {
    *uu = a.u[idx];  // closed-axis halo: pass-through
    *vv = a.v[idx];
  } else {
    *uu = a.fe[idx];  // staged (interior) or exchanged (remote halo)
    *vv = a.fn[idx];
  }
}

// Tiling: TJ=14 output rows x 60 output columns per 256-thread block,
// tile (TJ+2) x 64: the LDS fill is exactly 2 full 256-thread rounds
// and redundant update compute is 16/14 x 64/60 = 1.22x (vs 3x for a
// 1-row tile).  The fast kernel is PURE vector code — every scalar/
// boundary fallback lives in the ring kernels below, keeping the fast
// kernel at stage 19's VGPR budget (70 -> 7 waves/SIMD).  Store
// partition between the kernels is exact and index-only: dn*/h2 by
// stage19_pair_fast, u2/v2 by s30_fast_out.
constexpr int kTJ30 = 14;    // output rows per block
constexpr int kPC30 = 30;    // output pairs (2 cols) per row
constexpr int kTR30 = kTJ30 + 2;           // tile rows
constexpr int kTP30 = kPC30 + 2;           // tile pairs per row
constexpr int kFill30 = kTR30 * kTP30;     // LDS fill tasks

// friction output (j, pair i0) is servable from the fast kernel iff the
// nine (row, pair) LDS inputs it needs are all stage19-fast and the pair
// is stage27-fast; folds to pure bounds:
__device__ inline bool s30_fast_out(const SwArgs<float>& a, int j,
                                    int i0) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int jmin = a.f.south_open ? 1 : 2;
  const int jmax = (a.f.north_open && !a.f.north_wall) ? ny - 2 : ny - 3;
  const int imin = a.f.west_open ? 1 : 2;
  const int imax = (a.f.east_open && !a.f.east_wall) ? nx - 1 : nx - 2;
  return j - 1 >= jmin && j + 1 <= jmax && i0 - 2 >= imin &&
         i0 + 4 <= imax && i0 + 5 < nx;
}

template <int NT_>
__global__ void __launch_bounds__(NT_) sw_stage30t(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int bpc = (nx + 2 * kPC30 - 1) / (2 * kPC30);  // blocks per band
  const int T_ = (int)gridDim.x;
  const int b_ = (int)blockIdx.x;
  const int q8_ = T_ / 8, r8_ = T_ % 8, xc_ = b_ % 8, yc_ = b_ / 8;
  const int bid_ =
      (xc_ < r8_ ? xc_ * (q8_ + 1) : r8_ * (q8_ + 1) + (xc_ - r8_) * q8_)
      + yc_;
  const int jb = (bid_ / bpc) * kTJ30;       // first owned output row
  const int cb = (bid_ % bpc) * 2 * kPC30;   // first owned output column
  const int tid = (int)threadIdx.x;

  __shared__ float su[kTR30][2 * kTP30], sv[kTR30][2 * kTP30];
  for (int t = tid; t < kFill30; t += NT_) {
    const int lr = t / kTP30, lp = t % kTP30;
    const int rr = jb - 1 + lr;
    const int i0 = cb - 2 + 2 * lp;
    if (rr < 0 || !stage19_pair_fast(a, rr, i0)) continue;
    Upd2 w = stage19_math<false>(a, rr, i0);
    su[lr][2 * lp] = w.uu.x;
    su[lr][2 * lp + 1] = w.uu.y;
    sv[lr][2 * lp] = w.vv.x;
    sv[lr][2 * lp + 1] = w.vv.y;
    if (lr >= 1 && lr <= kTJ30 && lp >= 1 && lp <= kPC30) {
      const long long idx = (long long)rr * nx + i0;  // owned pair
      st2(a.dnh, idx, w.dnh);
      st2(a.dnu, idx, w.dnu);
      st2(a.dnv, idx, w.dnv);
      st2(a.h2, idx, w.hh);
      if (s30_in_ring(a, rr, i0)) {  // stage u'/v' for the ring friction
        st2(a.fe, idx, w.uu);
        st2(a.fn, idx, w.vv);
      }
    }
  }
  __syncthreads();

  for (int t = tid; t < kTJ30 * kPC30; t += NT_) {
    const int ro = t / kPC30, po = t % kPC30;
    const int j = jb + ro;
    const int i0 = cb + 2 * po;
    if (!s30_fast_out(a, j, i0)) continue;  // ring kernels own this pair
    const int lr = ro + 1, x = 2 * po + 2;
    const long long idx = (long long)j * nx + i0;
    // stage27_pair math, inputs from LDS
    vf2 uw = {su[lr][x - 1], su[lr][x]}, uc = {su[lr][x], su[lr][x + 1]},
        ue = {su[lr][x + 1], su[lr][x + 2]};
    vf2 un = {su[lr + 1][x], su[lr + 1][x + 1]},
        us = {su[lr - 1][x], su[lr - 1][x + 1]};
    vf2 vw = {sv[lr][x - 1], sv[lr][x]}, vcc = {sv[lr][x], sv[lr][x + 1]},
        ve = {sv[lr][x + 1], sv[lr][x + 2]};
    vf2 vn = {sv[lr + 1][x], sv[lr + 1][x + 1]},
        vs = {sv[lr - 1][x], sv[lr - 1][x + 1]};
    const float nu = a.nu;
    vf2 lu = (nu * (ue - uc) * a.rdx - nu * (uc - uw) * a.rdx) * a.rdx +
             (nu * (un - uc) * a.rdy - nu * (uc - us) * a.rdy) * a.rdy;
    vf2 lv = (nu * (ve - vcc) * a.rdx - nu * (vcc - vw) * a.rdx) * a.rdx +
             (nu * (vn - vcc) * a.rdy - nu * (vcc - vs) * a.rdy) * a.rdy;
    st2(a.u2, idx, uc + a.dt * lu);
    st2(a.v2, idx, vcc + a.dt * lv);
  }
}

// Ring kernels: every output cell the fast kernel skips — a ~3-wide
// perimeter.  Cells are enumerated as a compact ring (8 full edge rows +
// 12 edge columns of each middle row); the predicates, not the
// enumeration, define ownership, so over-coverage on tiny domains just
// re-writes identical values.  ringA does the update outputs (and
// stages u'/v' into fe/fn for the not-stage19-fast cells); ringB, a
// separate kernel so the strip is complete, does the friction from the
// staged strip.
__device__ inline bool s30_ring_cell(const SwArgs<float>& a, long long t,
                                     int* jo, int* io) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  const int edge_rows = ny < 8 ? ny : 8;
  const int edge_cols = nx < 12 ? nx : 12;
  const long long top = (long long)edge_rows * nx;
  const int mid = ny > 8 ? ny - 8 : 0;
  if (t >= top + (long long)mid * edge_cols) return false;
  if (t < top) {
    const int r = (int)(t / nx);
    // ny < 8: edge_rows == ny and r IS the row (the two-band formula
    // would alias low rows and skip the top ones)
    *jo = (r < 4 || ny < 8) ? r : ny - 8 + r;
    *io = (int)(t % nx);
  } else {
    const long long u = t - top;
    *jo = 4 + (int)(u / edge_cols);
    const int c = (int)(u % edge_cols);
    *io = (c < 6 || nx < 12) ? c : nx - 12 + c;
  }
  return true;
}

__global__ void sw_stage30_ringA(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  int j, i;
  if (!s30_ring_cell(a, (long long)blockIdx.x * blockDim.x + threadIdx.x,
                     &j, &i))
    return;
  const long long idx = (long long)j * nx + i;
  if (stage19_pair_fast(a, j, i & ~1)) return;  // fast kernel owns it
  if (j < 1 || j > ny - 2 || i < 1 || i > nx - 2) {
    a.h2[idx] = a.h[idx];
    return;
  }
  CellUpd<float> r8 = stage8_math(a, j, i);
  a.dnh[idx] = r8.dnh;
  a.dnu[idx] = r8.dnu;
  a.dnv[idx] = r8.dnv;
  a.h2[idx] = r8.hh;
  a.fe[idx] = r8.uu;  // complete the u'/v' strip for ringB
  a.fn[idx] = r8.vv;
}

// the stage7_cell friction (division math) for one interior ring cell,
// inputs from the staged u'/v' strip — used for a cell's own final AND
// recomputed (bitwise-identically) when that cell is the wrap source of
// a halo cell, so ringB can write the end-of-step halo refresh itself
__device__ inline void ringb_final_uv(const SwArgs<float>& a, int j,
                                      int i, float* uo, float* vo) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  float uc, vc, un, vn, us, vs, ue, ve, uw, vw;
  s30_read_uv(a, j, i, &uc, &vc);
  s30_read_uv(a, j + 1, i, &un, &vn);
  s30_read_uv(a, j - 1, i, &us, &vs);
  s30_read_uv(a, j, i + 1, &ue, &ve);
  s30_read_uv(a, j, i - 1, &uw, &vw);
  auto gu = [&](float c, float e, int ii) -> float {
    if (ii > nx - 2 || (ii == 0 && !a.f.west_open)) return 0.f;
    if (a.f.east_wall && ii == nx - 2) return 0.f;
    return a.nu * (e - c) / a.dx;
  };
  auto gv = [&](float c, float n, int jj) -> float {
    if (jj > ny - 2 || (jj == 0 && !a.f.south_open)) return 0.f;
    if (a.f.north_wall && jj == ny - 2) return 0.f;
    return a.nu * (n - c) / a.dy;
  };
  const float lu = (gu(uc, ue, i) - gu(uw, uc, i - 1)) / a.dx +
                   (gv(uc, un, j) - gv(us, uc, j - 1)) / a.dy;
  const float lv = (gu(vc, ve, i) - gu(vw, vc, i - 1)) / a.dx +
                   (gv(vc, vn, j) - gv(vs, vc, j - 1)) / a.dy;
  float uu2 = uc + a.dt * lu;
  float vv2 = vc + a.dt * lv;
  if (a.f.east_wall && i == nx - 2) uu2 = 0.f;
  if (a.f.north_wall && j == ny - 2) vv2 = 0.f;
  *uo = uu2;
  *vo = vv2;
}

__global__ void sw_stage30_ringB(SwArgs<float> a) {
  const int ny = (int)a.ny, nx = (int)a.nx;
  int j, i;
  if (!s30_ring_cell(a, (long long)blockIdx.x * blockDim.x + threadIdx.x,
                     &j, &i))
    return;
  const long long idx = (long long)j * nx + i;
  if (s30_fast_out(a, j, i & ~1)) return;  // fast kernel owns it
  if (j < 1 || j > ny - 2 || i < 1 || i > nx - 2) {
    // halo cell.  LOCAL periodic x halos (x_wrap): write what the
    // end-of-step wrap would leave, so the world-1 fused step needs NO
    // separate wrap/pack launches.  Remote-open halos: pass-through —
    // the real end-of-step exchange overwrites them.  Closed halos:
    // pass-through is the final value.  A wrap source that is itself a
    // (closed-y) halo cell passes the pre-update field through.
    int is = i;
    if (a.f.x_wrap) {
      if (i == 0) is = nx - 2;
      else if (i == nx - 1) is = 1;
    }
    const long long sdx = (long long)j * nx + is;
    if (is == i || j < 1 || j > ny - 2) {
      a.h2[idx] = a.h[sdx];  // pass-through (possibly x-wrapped corner)
      a.u2[idx] = a.u[sdx];
      a.v2[idx] = a.v[sdx];
    } else {
      a.h2[idx] = a.h2[sdx];  // update output, stored by a prior kernel
      // the wrap source is never s30_fast_out (cols 1/nx-2 fail its
      // bounds), so recompute its ringB final — bitwise-identical to
      // what its own thread stores
      float uu2, vv2;
      ringb_final_uv(a, j, is, &uu2, &vv2);
      a.u2[idx] = uu2;
      a.v2[idx] = vv2;
    }
    return;
  }
  float uu2, vv2;
  ringb_final_uv(a, j, i, &uu2, &vv2);
  a.u2[idx] = uu2;
  a.v2[idx] = vv2;
}


int sw_grid(long long n) {
  long long blocks = (n + kBlock - 1) / kBlock;
  if (blocks > 4096) blocks = 4096;  // grid-stride beyond this
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

}  // namespace

template <typename T>
static void sw_fill_args(SwArgs<T>& a, const SwLaunchParams& p) {
  a.fe = (T*)p.fe;
  a.fn = (T*)p.fn;
  a.q = (T*)p.q;
  a.ke = (T*)p.ke;
  a.h = (T*)p.h;
  a.u = (T*)p.u;
  a.v = (T*)p.v;
  a.dnh = (T*)p.dnh;
  a.dnu = (T*)p.dnu;
  a.dnv = (T*)p.dnv;
  a.doh = (T*)p.doh;
  a.dou = (T*)p.dou;
  a.dov = (T*)p.dov;
  a.h2 = (T*)p.h2;
  a.u2 = (T*)p.u2;
  a.v2 = (T*)p.v2;
  a.ny = p.ny;
  a.nx = p.nx;
  a.dx = (T)p.dx;
  a.dy = (T)p.dy;
  a.dt = (T)p.dt;
  a.nu = (T)p.nu;
  a.rdx = T(1) / (T)p.dx;
  a.rdy = T(1) / (T)p.dy;
  a.cor_base = (T)p.cor_base;
  a.cor_dj = (T)p.cor_dj;
  a.ab_a = (T)p.ab_a;
  a.ab_b = (T)p.ab_b;
  a.f = {p.south_open, p.north_open, p.west_open, p.east_open, p.east_wall,
         p.north_wall, p.x_wrap};
}

template <typename T>
static void sw_launch(int stage, const SwLaunchParams& p,
                      hipStream_t stream) {
  SwArgs<T> a;
  sw_fill_args(a, p);
  // one cell per thread; 1-D grid of row-major column-chunk tiles with
  // the XCD-aware remap applied inside the kernel (SW_BLOCK_MAP)
  long long gx = (p.nx + kBlock - 1) / kBlock;
  dim3 grid((unsigned)(gx * p.ny)), block(kBlock);
  switch (stage) {
    case 1: hipLaunchKernelGGL(sw_stage1_kernel<T>, grid, block, 0, stream, a); break;
    case 6: hipLaunchKernelGGL(sw_stage6_kernel<T>, grid, block, 0, stream, a); break;
    case 7: hipLaunchKernelGGL(sw_stage7_kernel<T>, grid, block, 0, stream, a); break;
    case 8: hipLaunchKernelGGL(sw_stage8_kernel<T>, grid, block, 0, stream, a); break;
    default: break;
  }
}

void launch_sw_stage(int stage, const SwLaunchParams& p, int is_double,
                     hipStream_t stream) {
  if (!is_double && stage >= 11) {
    // vectorized float stages: 11 -> stage1v, 16 -> stage6v, 17 -> stage7v
    SwArgs<float> a;
    sw_fill_args(a, p);
    long long ppr = stage >= 19 ? (p.nx + 1) / 2 : (p.nx + 3) / 4;
    long long packs = ppr * p.ny;
    dim3 grid((unsigned)((packs + 255) / 256)), block(256);
    if (stage >= 21 && stage <= 23) {  // band-tiled: TJ = 4 / 8 / 16
      int tjv = stage == 21 ? 4 : stage == 22 ? 8 : 16;
      long long tpc = 256 / tjv;
      long long tiles = ((p.ny + tjv - 1) / tjv) * ((ppr + tpc - 1) / tpc);
      grid = dim3((unsigned)tiles);
    }
    switch (stage) {
      case 11: hipLaunchKernelGGL(sw_stage1v, grid, block, 0, stream, a); break;
      case 16: hipLaunchKernelGGL(sw_stage6v, grid, block, 0, stream, a); break;
      case 17: hipLaunchKernelGGL(sw_stage7v, grid, block, 0, stream, a); break;
      case 18: hipLaunchKernelGGL(sw_stage18v, grid, block, 0, stream, a); break;
      case 19: hipLaunchKernelGGL(sw_stage19t<false>, grid, block, 0, stream, a); break;
      case 20: hipLaunchKernelGGL(sw_stage19t<true>, grid, block, 0, stream, a); break;
      case 21: hipLaunchKernelGGL(sw_stage21t<4>, grid, block, 0, stream, a); break;
      case 22: hipLaunchKernelGGL(sw_stage21t<8>, grid, block, 0, stream, a); break;
      case 23: hipLaunchKernelGGL(sw_stage21t<16>, grid, block, 0, stream, a); break;
      case 27: hipLaunchKernelGGL(sw_stage27v, grid, block, 0, stream, a); break;
      case 30:
      case 31:    // 31 = 512-thread blocks (single-round LDS fill)
      case 32: {  // 32 = fast + ringA only (N>1: a real fe/fn halo
                  // exchange runs before ringB, launched as stage 33)
        long long bpc = (p.nx + 2 * kPC30 - 1) / (2 * kPC30);
        long long bands = (p.ny + kTJ30 - 1) / kTJ30;
        dim3 g30((unsigned)(bpc * bands));
        if (stage == 31) {
          hipLaunchKernelGGL(sw_stage30t<512>, g30, dim3(512), 0, stream,
                             a);
        } else {
          hipLaunchKernelGGL(sw_stage30t<256>, g30, block, 0, stream, a);
        }
        long long er = p.ny < 8 ? p.ny : 8, ec = p.nx < 12 ? p.nx : 12;
        long long ring = er * p.nx +
                         (p.ny > 8 ? (p.ny - 8) * ec : 0);
        dim3 gcl((unsigned)((ring + 255) / 256));
        hipLaunchKernelGGL(sw_stage30_ringA, gcl, block, 0, stream, a);
        if (stage != 32) {
          hipLaunchKernelGGL(sw_stage30_ringB, gcl, block, 0, stream, a);
        }
        break;
      }
      case 33: {  // ringB alone, after the fe/fn halo exchange
        long long er = p.ny < 8 ? p.ny : 8, ec = p.nx < 12 ? p.nx : 12;
        long long ring = er * p.nx +
                         (p.ny > 8 ? (p.ny - 8) * ec : 0);
        dim3 gcl((unsigned)((ring + 255) / 256));
        hipLaunchKernelGGL(sw_stage30_ringB, gcl, block, 0, stream, a);
        break;
      }
      case 28: hipLaunchKernelGGL(sw_stage28v, grid, block, 0, stream, a); break;
      case 29: {
        long long ppr2 = (p.nx + 1) / 2;
        long long mid = p.ny - 5 > 0 ? p.ny - 5 : 0;
        long long ring = 5 * ppr2 + mid * 4;
        dim3 rg((unsigned)((ring + 255) / 256));
        hipLaunchKernelGGL(sw_stage29v, rg, block, 0, stream, a);
        break;
      }
    }
    return;
  }
  if (is_double) {
    sw_launch<double>(stage >= 11 ? stage - 10 : stage, p, stream);
  } else {
    sw_launch<float>(stage, p, stream);
  }
}

// ---------------------------------------------------------------- halo ops
// One kernel per exchange phase instead of torch narrow copies (measured
// ~110 us/step of copyBuffer/elementwise time in the eager halo path).

namespace {

template <typename T>
struct HaloArgs {
  T* f0;
  T* f1;
  T* f2;
  int nf, ny, nx;
  int col;  // for pack/unpack
};

template <typename T>
__device__ inline T* halo_field(const HaloArgs<T>& a, int f) {
  return f == 0 ? a.f0 : (f == 1 ? a.f1 : a.f2);
}

// periodic self-wrap, side 0: east halo col nx-1 <- col 1 ("send west");
// side 1: west halo col 0 <- col nx-2 ("send east")
template <typename T>
__global__ void halo_wrap_kernel(HaloArgs<T> a, int side) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= a.nf * a.ny) return;
  int f = t / a.ny, j = t % a.ny;
  T* p = halo_field(a, f) + (long long)j * a.nx;
  if (side != 1) p[a.nx - 1] = p[1];
  if (side != 0) p[0] = p[a.nx - 2];  // side 2 = both wraps in one launch
}

template <typename T>
__global__ void pack_cols_kernel(HaloArgs<T> a, T* buf) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= a.nf * a.ny) return;
  int f = t / a.ny, j = t % a.ny;
  buf[t] = halo_field(a, f)[(long long)j * a.nx + a.col];
}

template <typename T>
__global__ void unpack_cols_kernel(HaloArgs<T> a, const T* buf) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= a.nf * a.ny) return;
  int f = t / a.ny, j = t % a.ny;
  halo_field(a, f)[(long long)j * a.nx + a.col] = buf[t];
}

template <typename T>
HaloArgs<T> make_halo_args(void* const* fields, int nf, long long ny,
                           long long nx, long long col) {
  HaloArgs<T> a;
  a.f0 = (T*)fields[0];
  a.f1 = nf > 1 ? (T*)fields[1] : (T*)fields[0];
  a.f2 = nf > 2 ? (T*)fields[2] : (T*)fields[0];
  a.nf = nf;
  a.ny = (int)ny;
  a.nx = (int)nx;
  a.col = (int)col;
  return a;
}

int halo_grid(int n) { return (n + kBlock - 1) / kBlock; }

}  // namespace

void launch_halo_wrap(void* const* fields, int nf, long long ny,
                      long long nx, int side, int is_double,
                      hipStream_t stream) {
  int n = nf * (int)ny;
  if (is_double) {
    auto a = make_halo_args<double>(fields, nf, ny, nx, 0);
    hipLaunchKernelGGL(halo_wrap_kernel<double>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a, side);
  } else {
    auto a = make_halo_args<float>(fields, nf, ny, nx, 0);
    hipLaunchKernelGGL(halo_wrap_kernel<float>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a, side);
  }
}

void launch_pack_cols(void* buf, void* const* fields, int nf, long long ny,
                      long long nx, long long col, int is_double,
                      hipStream_t stream) {
  int n = nf * (int)ny;
  if (is_double) {
    auto a = make_halo_args<double>(fields, nf, ny, nx, col);
    hipLaunchKernelGGL(pack_cols_kernel<double>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a, (double*)buf);
  } else {
    auto a = make_halo_args<float>(fields, nf, ny, nx, col);
    hipLaunchKernelGGL(pack_cols_kernel<float>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a, (float*)buf);
  }
}

void launch_unpack_cols(void* const* fields, const void* buf, int nf,
                        long long ny, long long nx, long long col,
                        int is_double, hipStream_t stream) {
  int n = nf * (int)ny;
  if (is_double) {
    auto a = make_halo_args<double>(fields, nf, ny, nx, col);
    hipLaunchKernelGGL(unpack_cols_kernel<double>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a, (const double*)buf);
  } else {
    auto a = make_halo_args<float>(fields, nf, ny, nx, col);
    hipLaunchKernelGGL(unpack_cols_kernel<float>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a, (const float*)buf);
  }
}

// corner pack/unpack for the single-group halo exchange: buf layout is
// [diag d][field f] with d: 0=to-SW/from-NE, 1=to-SE/from-NW,
// 2=to-NW/from-SE, 3=to-NE/from-SW (matching parallel/grid.py halo_plan).
namespace {

template <typename T>
__global__ void pack_corners_kernel(HaloArgs<T> a, T* buf) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= 4 * a.nf) return;
  int d = t / a.nf, f = t % a.nf;
  const int ny = a.ny, nx = a.nx;
  int j = (d < 2) ? 1 : ny - 2;
  int i = (d == 0 || d == 2) ? 1 : nx - 2;
  buf[t] = halo_field(a, f)[(long long)j * nx + i];
}

template <typename T>
__global__ void unpack_corners_kernel(HaloArgs<T> a, const T* buf,
                                      int mask) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= 4 * a.nf) return;
  int d = t / a.nf, f = t % a.nf;
  if (!(mask & (1 << d))) return;
  const int ny = a.ny, nx = a.nx;
  // recv cells: d0 (ny-1,nx-1), d1 (ny-1,0), d2 (0,nx-1), d3 (0,0)
  int j = (d < 2) ? ny - 1 : 0;
  int i = (d == 0 || d == 2) ? nx - 1 : 0;
  halo_field(a, f)[(long long)j * nx + i] = buf[t];
}

// merged staging: wrap + both column packs + corner pack in ONE launch
// (work layout [wrap | cb0 | cb1 | corners]); see kernels.h for the
// contract.  All segments touch disjoint cells: wrap writes halo cols
// 0/nx-1, col packs read interior cols (1/nx-2), corner pack reads
// interior corner cells.
template <typename T>
struct HaloXArgs {
  T* f0;
  T* f1;
  T* f2;
  int nf, ny, nx;
  T* cb0;
  T* cb1;
  T* cor;
  int c0, c1;
  int wrap_side;  // -1 none, 0 east<-1, 1 west<-nx-2, 2 both
  int cor_mask;
};

template <typename T>
__device__ inline T* hx_field(const HaloXArgs<T>& a, int f) {
  return f == 0 ? a.f0 : (f == 1 ? a.f1 : a.f2);
}

template <typename T>
__global__ void pack_halo_kernel(HaloXArgs<T> a) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  const int ncol = a.nf * a.ny;
  if (t < ncol) {
    if (a.wrap_side < 0) return;
    int f = t / a.ny, j = t % a.ny;
    T* p = hx_field(a, f) + (long long)j * a.nx;
    if (a.wrap_side != 1) p[a.nx - 1] = p[1];
    if (a.wrap_side != 0) p[0] = p[a.nx - 2];
    return;
  }
  t -= ncol;
  if (t < 2 * ncol) {
    T* buf = t < ncol ? a.cb0 : a.cb1;
    int col = t < ncol ? a.c0 : a.c1;
    if (!buf) return;
    int s = t % ncol;
    int f = s / a.ny, j = s % a.ny;
    buf[s] = hx_field(a, f)[(long long)j * a.nx + col];
    return;
  }
  t -= 2 * ncol;
  if (t >= 4 * a.nf || !a.cor) return;
  int d = t / a.nf, f = t % a.nf;
  int j = (d < 2) ? 1 : a.ny - 2;
  int i = (d == 0 || d == 2) ? 1 : a.nx - 2;
  a.cor[t] = hx_field(a, f)[(long long)j * a.nx + i];
}

// which corner diagonal writes cell (j in {0, ny-1}, col in {0, nx-1}):
// d0 (ny-1,nx-1), d1 (ny-1,0), d2 (0,nx-1), d3 (0,0)
__device__ inline int corner_diag(bool top_row, bool east_col) {
  return top_row ? (east_col ? 0 : 1) : (east_col ? 2 : 3);
}

template <typename T>
__global__ void unpack_halo_kernel(HaloXArgs<T> a) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  const int ncol = a.nf * a.ny;
  if (t < 2 * ncol) {
    const T* buf = t < ncol ? a.cb0 : a.cb1;
    int col = t < ncol ? a.c0 : a.c1;
    if (!buf) return;
    int s = t % ncol;
    int f = s / a.ny, j = s % a.ny;
    if ((j == 0 || j == a.ny - 1) && a.cor) {
      // the corner segment of this same launch owns this cell
      int d = corner_diag(j == a.ny - 1, col == a.nx - 1);
      if (a.cor_mask & (1 << d)) return;
    }
    hx_field(a, f)[(long long)j * a.nx + col] = buf[s];
    return;
  }
  t -= 2 * ncol;
  if (t >= 4 * a.nf || !a.cor) return;
  int d = t / a.nf, f = t % a.nf;
  if (!(a.cor_mask & (1 << d))) return;
  int j = (d < 2) ? a.ny - 1 : 0;
  int i = (d == 0 || d == 2) ? a.nx - 1 : 0;
  hx_field(a, f)[(long long)j * a.nx + i] = a.cor[t];
}

template <typename T>
HaloXArgs<T> make_hx_args(void* const* fields, int nf, long long ny,
                          long long nx, int wrap_side, void* cb0,
                          long long c0, void* cb1, long long c1, void* cor,
                          int cor_mask) {
  HaloXArgs<T> a;
  a.f0 = (T*)fields[0];
  a.f1 = nf > 1 ? (T*)fields[1] : (T*)fields[0];
  a.f2 = nf > 2 ? (T*)fields[2] : (T*)fields[0];
  a.nf = nf;
  a.ny = (int)ny;
  a.nx = (int)nx;
  a.cb0 = (T*)cb0;
  a.cb1 = (T*)cb1;
  a.cor = (T*)cor;
  a.c0 = (int)c0;
  a.c1 = (int)c1;
  a.wrap_side = wrap_side;
  a.cor_mask = cor_mask;
  return a;
}

}  // namespace

void launch_pack_corners(void* buf, void* const* fields, int nf,
                         long long ny, long long nx, int is_double,
                         hipStream_t stream) {
  int n = 4 * nf;
  if (is_double) {
    auto a = make_halo_args<double>(fields, nf, ny, nx, 0);
    hipLaunchKernelGGL(pack_corners_kernel<double>, dim3(1), dim3(64), 0,
                       stream, a, (double*)buf);
  } else {
    auto a = make_halo_args<float>(fields, nf, ny, nx, 0);
    hipLaunchKernelGGL(pack_corners_kernel<float>, dim3(1), dim3(64), 0,
                       stream, a, (float*)buf);
  }
  (void)n;
}

void launch_unpack_corners(void* const* fields, const void* buf, int nf,
                           long long ny, long long nx, int mask,
                           int is_double, hipStream_t stream) {
  if (is_double) {
    auto a = make_halo_args<double>(fields, nf, ny, nx, 0);
    hipLaunchKernelGGL(unpack_corners_kernel<double>, dim3(1), dim3(64), 0,
                       stream, a, (const double*)buf, mask);
  } else {
    auto a = make_halo_args<float>(fields, nf, ny, nx, 0);
    hipLaunchKernelGGL(unpack_corners_kernel<float>, dim3(1), dim3(64), 0,
                       stream, a, (const float*)buf, mask);
  }
}

void launch_pack_halo(void* const* fields, int nf, long long ny,
                      long long nx, int wrap_side, void* cb0, long long c0,
                      void* cb1, long long c1, void* cor, int is_double,
                      hipStream_t stream) {
  int n = 3 * nf * (int)ny + 4 * nf;
  if (is_double) {
    auto a = make_hx_args<double>(fields, nf, ny, nx, wrap_side, cb0, c0,
                                  cb1, c1, cor, 0);
    hipLaunchKernelGGL(pack_halo_kernel<double>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a);
  } else {
    auto a = make_hx_args<float>(fields, nf, ny, nx, wrap_side, cb0, c0,
                                 cb1, c1, cor, 0);
    hipLaunchKernelGGL(pack_halo_kernel<float>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a);
  }
}

void launch_unpack_halo(void* const* fields, int nf, long long ny,
                        long long nx, void* cb0, long long c0, void* cb1,
                        long long c1, void* cor, int cor_mask,
                        int is_double, hipStream_t stream) {
  int n = 2 * nf * (int)ny + 4 * nf;
  if (is_double) {
    auto a = make_hx_args<double>(fields, nf, ny, nx, -1, cb0, c0, cb1, c1,
                                  cor, cor_mask);
    hipLaunchKernelGGL(unpack_halo_kernel<double>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a);
  } else {
    auto a = make_hx_args<float>(fields, nf, ny, nx, -1, cb0, c0, cb1, c1,
                                 cor, cor_mask);
    hipLaunchKernelGGL(unpack_halo_kernel<float>, dim3(halo_grid(n)),
                       dim3(kBlock), 0, stream, a);
  }
}
