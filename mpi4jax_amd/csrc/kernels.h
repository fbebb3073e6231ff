// CDNA4 device kernels (gfx950) — C ABI, torch-free.
#pragma once

#include <hip/hip_runtime.h>

// dtype codes shared between bridge.cpp and kernels.hip
enum DtCode : int {
  DT_F32 = 0,
  DT_F64 = 1,
  DT_F16 = 2,
  DT_BF16 = 3,
  DT_I8 = 4,
  DT_U8 = 5,
  DT_I32 = 6,
  DT_I64 = 7,
};

enum OpCode : int {  // matches ncclRedOp_t for the first four
  OPC_SUM = 0,
  OPC_PROD = 1,
  OPC_MAX = 2,
  OPC_MIN = 3,
  // kernel-only codes (never passed to an RCCL collective — RCCL has no
  // bitwise reductions; these ride the p2p+combine compositions)
  OPC_BAND = 5,
  OPC_BOR = 6,
  OPC_BXOR = 7,
};

// dst[i] = a[i] (op) b[i], grid-strided, enqueued on `stream`
void launch_combine(void* dst, const void* a, const void* b, long long n,
                    int dt_code, int op_code, hipStream_t stream);

// out[r*cols + c] = in[r*stride0 + c*stride1]   (strides in ELEMENTS)
// LDS-staged tiles when neither axis is contiguous on the write side.
void launch_pack2d(void* out, const void* in, long long rows, long long cols,
                   long long stride0, long long stride1, int elem_size,
                   hipStream_t stream);

// in is contiguous (rows, cols); out[r*stride0 + c*stride1] = in[r*cols + c]
void launch_unpack2d(void* out, const void* in, long long rows,
                     long long cols, long long stride0, long long stride1,
                     int elem_size, hipStream_t stream);

// fused shallow-water step stages (shallow_water.hip)
struct SwLaunchParams {
  void *fe, *fn, *q, *ke;
  void *h, *u, *v;
  void *dnh, *dnu, *dnv;
  void *doh, *dou, *dov;
  void *h2, *u2, *v2;  // double-buffer outputs for the merged stages
  long long ny, nx;
  double dx, dy, dt, nu;
  double cor_base, cor_dj;
  double ab_a, ab_b;
  int south_open, north_open, west_open, east_open;
  int east_wall, north_wall;
  int x_wrap;  // x halos are LOCAL periodic wraps (nproc_x==1), not remote
};

void launch_sw_stage(int stage, const SwLaunchParams& p, int is_double,
                     hipStream_t stream);

// halo-exchange helpers (shallow_water.hip): periodic wrap and multi-field
// column pack/unpack (up to 3 fields)
void launch_halo_wrap(void* const* fields, int nf, long long ny,
                      long long nx, int side, int is_double,
                      hipStream_t stream);
void launch_pack_cols(void* buf, void* const* fields, int nf, long long ny,
                      long long nx, long long col, int is_double,
                      hipStream_t stream);
void launch_unpack_cols(void* const* fields, const void* buf, int nf,
                        long long ny, long long nx, long long col,
                        int is_double, hipStream_t stream);

// corner pack/unpack for the single-group halo exchange
void launch_pack_corners(void* buf, void* const* fields, int nf,
                         long long ny, long long nx, int is_double,
                         hipStream_t stream);
void launch_unpack_corners(void* const* fields, const void* buf, int nf,
                           long long ny, long long nx, int mask,
                           int is_double, hipStream_t stream);

// merged halo staging: one launch covering the periodic wrap, both column
// packs and the corner pack (resp. the column + corner unpacks) — the
// per-exchange launch count is what bounds strong scaling once the local
// domain is small.  Null buffer => that segment is skipped; wrap_side -1
// => no wrap.  The unpack skips a column's j=0 / j=ny-1 cell when the
// corner segment of the SAME launch writes it (cor_mask bit set), which
// preserves the "corners win" ordering rule of parallel/grid.halo_plan.
void launch_pack_halo(void* const* fields, int nf, long long ny,
                      long long nx, int wrap_side, void* cb0, long long c0,
                      void* cb1, long long c1, void* cor, int is_double,
                      hipStream_t stream);
void launch_unpack_halo(void* const* fields, int nf, long long ny,
                        long long nx, void* cb0, long long c0, void* cb1,
                        long long c1, void* cor, int cor_mask,
                        int is_double, hipStream_t stream);
