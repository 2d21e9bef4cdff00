/* binop.hip — binary operators on the result grid.
 *
 * Independent MI355X-native implementation of the sample-level math in
 * app/vmselect/promql/binary_op.go + metricsql/binaryop/funcs.go.  Label
 * matching (createTimeseriesMapByTagSet / adjustBinaryOpTags /
 * groupJoin, binary_op.go:271-470) is host metadata work and stays in the
 * host mirror (victoriametrics_amd/binary_op.py); the kernels here do the
 * per-point arithmetic over the matched pair lists:
 *
 *   binop_pairs_kernel   newBinaryOpFunc's value loop (binary_op.go:162-236):
 *                        arith/cmp ops with bool modifier, fill_left/right
 *                        and the dropNaNRight comparison rule.
 *   binop_mask_kernel    and/if (addRightNaNsToLeft, :549), unless/ifnot
 *                        (addLeftNaNsIfNoRightNaNs, :729), default
 *                        (fillLeftNaNsWithRightValues, :622) — left rows
 *                        masked/filled against a CSR group of right rows.
 *   binop_or_kernel      or (fillLeftNaNsWithRightValuesOrMerge, :645):
 *                        the order-dependent left×right merge walk, one
 *                        thread per (group, grid point), canBeMerged
 *                        precomputed on the host from marshaled names.
 *
 * All ops keep Go float64 semantics bit-for-bit: Eq(NaN,NaN)=true,
 * Pow(NaN,_)=NaN, Mod=math.Mod=fmod, comparisons false on NaN.
 */
#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdio>
#include <vector>

#include "../../include/vmgpu.h"

namespace {

int bset_err(char* errbuf, size_t len, const char* msg) {
  if (errbuf && len) snprintf(errbuf, len, "%s", msg);
  return 1;
}

int bhip_err(char* errbuf, size_t len, const char* what, hipError_t e) {
  if (errbuf && len) snprintf(errbuf, len, "%s: %s", what, hipGetErrorString(e));
  return 2;
}

#define BHIP_TRY(expr, what)                                               \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) return bhip_err(errbuf, errbuf_len, what, _e);   \
  } while (0)

}  // namespace

static __device__ __forceinline__ double b_nan() {
  return __longlong_as_double(0x7ff8000000000000LL);
}

/* metricsql/binaryop/funcs.go, applied with the bool-modifier wrapper from
 * newBinaryOpCmpFunc (binary_op.go:136-153). */
/* x86-64 (and Go-on-amd64) SSE NaN propagation: the FIRST NaN operand's
 * payload, quieted.  CDNA VALU ops instead propagate through the negated/
 * canonicalized encoding (a-b lowers to add(a,-b), flipping a NaN b's sign
 * bit), so arith ops route NaNs explicitly to stay bit-identical with the
 * host path. */
static __device__ __forceinline__ double b_quiet(double x) {
  return __longlong_as_double(__double_as_longlong(x) |
                              0x0008000000000000LL);
}

#define B_ARITH(expr)                                      \
  do {                                                     \
    if (isnan(a)) return b_quiet(a);                       \
    if (isnan(b)) return b_quiet(b);                       \
    return (expr);                                         \
  } while (0)

static __device__ double binop_apply(int op, int is_bool, double a, double b) {
  bool cmp_hit = false;
  switch (op) {
    case VMGPU_BINOP_PLUS:  B_ARITH(a + b);
    case VMGPU_BINOP_MINUS: B_ARITH(a - b);
    case VMGPU_BINOP_MUL:   B_ARITH(a * b);
    case VMGPU_BINOP_DIV:   B_ARITH(a / b);
    case VMGPU_BINOP_MOD:   B_ARITH(fmod(a, b));
    case VMGPU_BINOP_POW:
      /* binaryop.Pow: NaN^any = NaN; then C99 pow edge cases */
      if (isnan(a)) return b_nan();
      if (isnan(b)) return (a == 1.0) ? 1.0 : b_quiet(b);
      return pow(a, b);
    case VMGPU_BINOP_ATAN2:
      if (isnan(a)) return b_quiet(a);
      if (isnan(b)) return b_quiet(b);
      return atan2(a, b);
    case VMGPU_BINOP_EQ:    cmp_hit = isnan(a) ? isnan(b) : (a == b); break;
    case VMGPU_BINOP_NEQ:
      cmp_hit = isnan(a) ? !isnan(b) : (isnan(b) ? true : (a != b));
      break;
    case VMGPU_BINOP_GT:  cmp_hit = a > b;  break;
    case VMGPU_BINOP_LT:  cmp_hit = a < b;  break;
    case VMGPU_BINOP_GTE: cmp_hit = a >= b; break;
    case VMGPU_BINOP_LTE: cmp_hit = a <= b; break;
    case VMGPU_BINOP_DEFAULT: return isnan(a) ? b : a;
    case VMGPU_BINOP_IF:      return isnan(b) ? b_nan() : a;
    case VMGPU_BINOP_IFNOT:   return isnan(b) ? a : b_nan();
    case VMGPU_BINOP_AND:     return (isnan(a) || isnan(b)) ? b_nan() : a;
    case VMGPU_BINOP_OR:      return !isnan(a) ? a : b;
    default: return b_nan();
  }
  if (!is_bool) return cmp_hit ? a : b_nan();
  if (isnan(a)) return b_nan();
  return cmp_hit ? 1.0 : 0.0;
}

__global__ void binop_pairs_kernel(int op, int is_bool, int drop_nan_right,
                                   const double* left_vals, const uint32_t* left_idx,
                                   const double* right_vals, const uint32_t* right_idx,
                                   uint32_t n_pairs, uint32_t n_grid,
                                   int has_fill_left, double fill_left,
                                   int has_fill_right, double fill_right,
                                   double* out) {
  uint64_t total = (uint64_t)n_pairs * n_grid;
  for (uint64_t e = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += (uint64_t)gridDim.x * blockDim.x) {
    uint32_t p = (uint32_t)(e / n_grid);
    uint32_t g = (uint32_t)(e % n_grid);
    uint32_t lr = left_idx ? left_idx[p] : p;
    uint32_t rr = right_idx ? right_idx[p] : p;
    double a = left_vals[(uint64_t)lr * n_grid + g];
    double b = right_vals[(uint64_t)rr * n_grid + g];
    bool ln = isnan(a), rn = isnan(b);
    double r;
    if (ln && rn) {
      r = binop_apply(op, is_bool, a, b);
    } else if (drop_nan_right && rn && !has_fill_right) {
      r = b_nan();
    } else {
      if (ln && has_fill_left) a = fill_left;
      if (rn && has_fill_right) b = fill_right;
      r = binop_apply(op, is_bool, a, b);
    }
    out[e] = r;
  }
}

#define VMGPU_MASK_AND 0     /* left=NaN where NO right in group has value */
#define VMGPU_MASK_UNLESS 1  /* left=NaN where ANY right in group has value */
#define VMGPU_MASK_DEFAULT 2 /* left NaN -> first non-NaN right in group */

__global__ void binop_mask_kernel(int mode, double* left_vals,
                                  const uint32_t* left_group, uint32_t n_left,
                                  const double* right_vals,
                                  const uint32_t* group_offsets, /* [n_groups+1] right rows CSR */
                                  const uint32_t* group_rows,
                                  uint32_t n_grid) {
  uint64_t total = (uint64_t)n_left * n_grid;
  for (uint64_t e = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += (uint64_t)gridDim.x * blockDim.x) {
    uint32_t l = (uint32_t)(e / n_grid);
    uint32_t g = (uint32_t)(e % n_grid);
    uint32_t grp = left_group[l];
    uint32_t lo = group_offsets[grp], hi = group_offsets[grp + 1];
    if (mode == VMGPU_MASK_DEFAULT) {
      if (!isnan(left_vals[e])) continue;
      for (uint32_t k = lo; k < hi; k++) {
        double v = right_vals[(uint64_t)group_rows[k] * n_grid + g];
        if (!isnan(v)) { left_vals[e] = v; break; }
      }
    } else {
      bool has = false;
      for (uint32_t k = lo; k < hi && !has; k++)
        has = !isnan(right_vals[(uint64_t)group_rows[k] * n_grid + g]);
      if (mode == VMGPU_MASK_AND ? !has : has) left_vals[e] = b_nan();
    }
  }
}

/* `or` merge walk (binary_op.go:645-700): per (group, grid point), the exact
 * Go loop order — lefts outer, rights inner; a right sample is consumed
 * (set to NaN) by the first left that either has a value or can merge with
 * it, and fills that left when the left is NaN and names match. */
__global__ void binop_or_kernel(double* left_vals, double* right_vals,
                                const uint32_t* lgroup_offsets, const uint32_t* lgroup_rows,
                                const uint32_t* rgroup_offsets, const uint32_t* rgroup_rows,
                                const uint8_t* can_merge, /* [sum(nl*nr)] per group */
                                const uint64_t* merge_offsets, /* [n_groups] base into can_merge */
                                uint32_t n_groups, uint32_t n_grid) {
  uint64_t total = (uint64_t)n_groups * n_grid;
  for (uint64_t e = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; e < total;
       e += (uint64_t)gridDim.x * blockDim.x) {
    uint32_t grp = (uint32_t)(e / n_grid);
    uint32_t g = (uint32_t)(e % n_grid);
    uint32_t ll = lgroup_offsets[grp], lh = lgroup_offsets[grp + 1];
    uint32_t rl = rgroup_offsets[grp], rh = rgroup_offsets[grp + 1];
    uint32_t nr = rh - rl;
    const uint8_t* cm = can_merge + merge_offsets[grp];
    for (uint32_t li = ll; li < lh; li++) {
      uint64_t le = (uint64_t)lgroup_rows[li] * n_grid + g;
      double lv = left_vals[le];
      bool left_nan = isnan(lv);
      for (uint32_t ri = rl; ri < rh; ri++) {
        uint64_t re = (uint64_t)rgroup_rows[ri] * n_grid + g;
        bool mergeable = cm[(uint64_t)(li - ll) * nr + (ri - rl)] != 0;
        double rv = right_vals[re];
        if (left_nan && mergeable) {
          left_vals[le] = rv;
          lv = rv;  /* Go reads valuesLeft[i] once per (l); fill persists */
        }
        if (!left_nan || mergeable) right_vals[re] = b_nan();
      }
    }
  }
}

namespace {

struct DevBuf {
  void* p = nullptr;
  ~DevBuf() { if (p) (void)hipFree(p); }
  hipError_t alloc(size_t n) { return hipMalloc(&p, n ? n : 1); }
};

}  // namespace

extern "C" {

int vmgpu_binop_eval(int32_t op, int32_t is_bool, int32_t drop_nan_right,
                     const double* left_vals, uint32_t n_left_rows,
                     const uint32_t* left_idx,
                     const double* right_vals, uint32_t n_right_rows,
                     const uint32_t* right_idx,
                     uint32_t n_pairs, uint32_t n_grid,
                     int32_t has_fill_left, double fill_left,
                     int32_t has_fill_right, double fill_right,
                     double* out, char* errbuf, size_t errbuf_len) {
  if (!left_vals || !right_vals || !out || n_pairs == 0 || n_grid == 0)
    return bset_err(errbuf, errbuf_len, "vmgpu: bad binop args");
  hipStream_t st = 0;
  size_t lbytes = (size_t)n_left_rows * n_grid * 8;
  size_t rbytes = (size_t)n_right_rows * n_grid * 8;
  size_t obytes = (size_t)n_pairs * n_grid * 8;
  DevBuf dl, dr, dli, dri, dout;
  BHIP_TRY(dl.alloc(lbytes), "alloc binop left");
  BHIP_TRY(dr.alloc(rbytes), "alloc binop right");
  BHIP_TRY(dout.alloc(obytes), "alloc binop out");
  BHIP_TRY(hipMemcpyAsync(dl.p, left_vals, lbytes, hipMemcpyHostToDevice, st), "ul left");
  BHIP_TRY(hipMemcpyAsync(dr.p, right_vals, rbytes, hipMemcpyHostToDevice, st), "ul right");
  uint32_t* d_li = nullptr;
  uint32_t* d_ri = nullptr;
  if (left_idx) {
    BHIP_TRY(dli.alloc((size_t)n_pairs * 4), "alloc left idx");
    BHIP_TRY(hipMemcpyAsync(dli.p, left_idx, (size_t)n_pairs * 4, hipMemcpyHostToDevice, st), "ul left idx");
    d_li = (uint32_t*)dli.p;
  }
  if (right_idx) {
    BHIP_TRY(dri.alloc((size_t)n_pairs * 4), "alloc right idx");
    BHIP_TRY(hipMemcpyAsync(dri.p, right_idx, (size_t)n_pairs * 4, hipMemcpyHostToDevice, st), "ul right idx");
    d_ri = (uint32_t*)dri.p;
  }
  uint64_t total = (uint64_t)n_pairs * n_grid;
  uint32_t blocks = (uint32_t)std::min<uint64_t>((total + 255) / 256, 4096);
  hipLaunchKernelGGL(binop_pairs_kernel, dim3(blocks), dim3(256), 0, st,
                     op, is_bool, drop_nan_right, (const double*)dl.p, d_li,
                     (const double*)dr.p, d_ri, n_pairs, n_grid,
                     has_fill_left, fill_left, has_fill_right, fill_right,
                     (double*)dout.p);
  BHIP_TRY(hipMemcpyAsync(out, dout.p, obytes, hipMemcpyDeviceToHost, st), "dl out");
  BHIP_TRY(hipStreamSynchronize(st), "sync binop");
  hipError_t kerr = hipGetLastError();
  if (kerr != hipSuccess) return bhip_err(errbuf, errbuf_len, "binop kernel", kerr);
  return 0;
}

int vmgpu_binop_mask(int32_t mode, double* left_vals, uint32_t n_left,
                     const uint32_t* left_group,
                     const double* right_vals, uint32_t n_right,
                     const uint32_t* group_offsets, uint32_t n_groups,
                     const uint32_t* group_rows,
                     uint32_t n_grid, char* errbuf, size_t errbuf_len) {
  if (!left_vals || !left_group || !right_vals || !group_offsets || !group_rows ||
      n_left == 0 || n_grid == 0)
    return bset_err(errbuf, errbuf_len, "vmgpu: bad binop mask args");
  hipStream_t st = 0;
  size_t lbytes = (size_t)n_left * n_grid * 8;
  size_t rbytes = (size_t)n_right * n_grid * 8;
  uint32_t n_rows = group_offsets[n_groups];
  DevBuf dl, dr, dlg, dgo, dgr;
  BHIP_TRY(dl.alloc(lbytes), "alloc mask left");
  BHIP_TRY(dr.alloc(rbytes), "alloc mask right");
  BHIP_TRY(dlg.alloc((size_t)n_left * 4), "alloc mask lgroup");
  BHIP_TRY(dgo.alloc((size_t)(n_groups + 1) * 4), "alloc mask goff");
  BHIP_TRY(dgr.alloc((size_t)n_rows * 4), "alloc mask grows");
  BHIP_TRY(hipMemcpyAsync(dl.p, left_vals, lbytes, hipMemcpyHostToDevice, st), "ul mask left");
  BHIP_TRY(hipMemcpyAsync(dr.p, right_vals, rbytes, hipMemcpyHostToDevice, st), "ul mask right");
  BHIP_TRY(hipMemcpyAsync(dlg.p, left_group, (size_t)n_left * 4, hipMemcpyHostToDevice, st), "ul mask lg");
  BHIP_TRY(hipMemcpyAsync(dgo.p, group_offsets, (size_t)(n_groups + 1) * 4, hipMemcpyHostToDevice, st), "ul mask go");
  BHIP_TRY(hipMemcpyAsync(dgr.p, group_rows, (size_t)n_rows * 4, hipMemcpyHostToDevice, st), "ul mask gr");
  uint64_t total = (uint64_t)n_left * n_grid;
  uint32_t blocks = (uint32_t)std::min<uint64_t>((total + 255) / 256, 4096);
  hipLaunchKernelGGL(binop_mask_kernel, dim3(blocks), dim3(256), 0, st,
                     mode, (double*)dl.p, (const uint32_t*)dlg.p, n_left,
                     (const double*)dr.p, (const uint32_t*)dgo.p,
                     (const uint32_t*)dgr.p, n_grid);
  BHIP_TRY(hipMemcpyAsync(left_vals, dl.p, lbytes, hipMemcpyDeviceToHost, st), "dl mask left");
  BHIP_TRY(hipStreamSynchronize(st), "sync mask");
  hipError_t kerr = hipGetLastError();
  if (kerr != hipSuccess) return bhip_err(errbuf, errbuf_len, "mask kernel", kerr);
  return 0;
}

int vmgpu_binop_or(double* left_vals, uint32_t n_left,
                   double* right_vals, uint32_t n_right,
                   const uint32_t* lgroup_offsets, const uint32_t* lgroup_rows,
                   const uint32_t* rgroup_offsets, const uint32_t* rgroup_rows,
                   const uint8_t* can_merge, const uint64_t* merge_offsets,
                   uint64_t merge_len, uint32_t n_groups, uint32_t n_grid,
                   char* errbuf, size_t errbuf_len) {
  if (!left_vals || !right_vals || n_groups == 0 || n_grid == 0)
    return bset_err(errbuf, errbuf_len, "vmgpu: bad binop or args");
  hipStream_t st = 0;
  size_t lbytes = (size_t)n_left * n_grid * 8;
  size_t rbytes = (size_t)n_right * n_grid * 8;
  uint32_t nlr = lgroup_offsets[n_groups];
  uint32_t nrr = rgroup_offsets[n_groups];
  DevBuf dl, dr, dlo, dlr, dro, drr, dcm, dmo;
  BHIP_TRY(dl.alloc(lbytes), "alloc or left");
  BHIP_TRY(dr.alloc(rbytes), "alloc or right");
  BHIP_TRY(dlo.alloc((size_t)(n_groups + 1) * 4), "alloc or lo");
  BHIP_TRY(dlr.alloc((size_t)nlr * 4), "alloc or lr");
  BHIP_TRY(dro.alloc((size_t)(n_groups + 1) * 4), "alloc or ro");
  BHIP_TRY(drr.alloc((size_t)nrr * 4), "alloc or rr");
  BHIP_TRY(dcm.alloc(merge_len), "alloc or cm");
  BHIP_TRY(dmo.alloc((size_t)n_groups * 8), "alloc or mo");
  BHIP_TRY(hipMemcpyAsync(dl.p, left_vals, lbytes, hipMemcpyHostToDevice, st), "ul or l");
  BHIP_TRY(hipMemcpyAsync(dr.p, right_vals, rbytes, hipMemcpyHostToDevice, st), "ul or r");
  BHIP_TRY(hipMemcpyAsync(dlo.p, lgroup_offsets, (size_t)(n_groups + 1) * 4, hipMemcpyHostToDevice, st), "ul or lo");
  BHIP_TRY(hipMemcpyAsync(dlr.p, lgroup_rows, (size_t)nlr * 4, hipMemcpyHostToDevice, st), "ul or lr");
  BHIP_TRY(hipMemcpyAsync(dro.p, rgroup_offsets, (size_t)(n_groups + 1) * 4, hipMemcpyHostToDevice, st), "ul or ro");
  BHIP_TRY(hipMemcpyAsync(drr.p, rgroup_rows, (size_t)nrr * 4, hipMemcpyHostToDevice, st), "ul or rr");
  BHIP_TRY(hipMemcpyAsync(dcm.p, can_merge, merge_len, hipMemcpyHostToDevice, st), "ul or cm");
  BHIP_TRY(hipMemcpyAsync(dmo.p, merge_offsets, (size_t)n_groups * 8, hipMemcpyHostToDevice, st), "ul or mo");
  uint64_t total = (uint64_t)n_groups * n_grid;
  uint32_t blocks = (uint32_t)std::min<uint64_t>((total + 255) / 256, 4096);
  hipLaunchKernelGGL(binop_or_kernel, dim3(blocks), dim3(256), 0, st,
                     (double*)dl.p, (double*)dr.p,
                     (const uint32_t*)dlo.p, (const uint32_t*)dlr.p,
                     (const uint32_t*)dro.p, (const uint32_t*)drr.p,
                     (const uint8_t*)dcm.p, (const uint64_t*)dmo.p,
                     n_groups, n_grid);
  BHIP_TRY(hipMemcpyAsync(left_vals, dl.p, lbytes, hipMemcpyDeviceToHost, st), "dl or l");
  BHIP_TRY(hipMemcpyAsync(right_vals, dr.p, rbytes, hipMemcpyDeviceToHost, st), "dl or r");
  BHIP_TRY(hipStreamSynchronize(st), "sync or");
  hipError_t kerr = hipGetLastError();
  if (kerr != hipSuccess) return bhip_err(errbuf, errbuf_len, "or kernel", kerr);
  return 0;
}

}  /* extern "C" */
