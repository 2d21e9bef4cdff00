/* oracle/binop.c — TEST INFRASTRUCTURE ONLY (never linked into the product
 * path; see oracle/__init__.py header).
 *
 * CPU restatement of the binary-operator sample math:
 *   - metricsql/binaryop/funcs.go (vendored at
 *     vendor/github.com/VictoriaMetrics/metricsql/binaryop/funcs.go):
 *     Eq/Neq NaN rules, Pow(NaN,_)=NaN, Mod=math.Mod, Default/If/Ifnot/
 *     And/Or.
 *   - newBinaryOpFunc's value loop (binary_op.go:162-236): bool modifier,
 *     fill_left/fill_right, dropNaNRight.
 * Go's math.Mod/math.Pow/math.Atan2 delegate to the C library functions
 * with identical IEEE-754 semantics, so fmod/pow/atan2 match bitwise.
 */
#include <math.h>
#include <stdint.h>

#define BINOP_PLUS 0
#define BINOP_MINUS 1
#define BINOP_MUL 2
#define BINOP_DIV 3
#define BINOP_MOD 4
#define BINOP_POW 5
#define BINOP_ATAN2 6
#define BINOP_EQ 7
#define BINOP_NEQ 8
#define BINOP_GT 9
#define BINOP_LT 10
#define BINOP_GTE 11
#define BINOP_LTE 12
#define BINOP_DEFAULT 13
#define BINOP_IF 14
#define BINOP_IFNOT 15
#define BINOP_AND 16
#define BINOP_OR 17

static double bnan(void) { return nan(""); }

double vm_binop_scalar(int32_t op, int32_t is_bool, double a, double b) {
  int cmp_hit;
  switch (op) {
    case BINOP_PLUS:  return a + b;
    case BINOP_MINUS: return a - b;
    case BINOP_MUL:   return a * b;
    case BINOP_DIV:   return a / b;
    case BINOP_MOD:   return fmod(a, b);
    case BINOP_POW:   return isnan(a) ? bnan() : pow(a, b);
    case BINOP_ATAN2: return atan2(a, b);
    case BINOP_EQ:    cmp_hit = isnan(a) ? isnan(b) : (a == b); break;
    case BINOP_NEQ:
      cmp_hit = isnan(a) ? !isnan(b) : (isnan(b) ? 1 : (a != b));
      break;
    case BINOP_GT:  cmp_hit = a > b;  break;
    case BINOP_LT:  cmp_hit = a < b;  break;
    case BINOP_GTE: cmp_hit = a >= b; break;
    case BINOP_LTE: cmp_hit = a <= b; break;
    case BINOP_DEFAULT: return isnan(a) ? b : a;
    case BINOP_IF:      return isnan(b) ? bnan() : a;
    case BINOP_IFNOT:   return isnan(b) ? a : bnan();
    case BINOP_AND:     return (isnan(a) || isnan(b)) ? bnan() : a;
    case BINOP_OR:      return !isnan(a) ? a : b;
    default: return bnan();
  }
  if (!is_bool) return cmp_hit ? a : bnan();
  if (isnan(a)) return bnan();
  return cmp_hit ? 1.0 : 0.0;
}

/* newBinaryOpFunc value loop over one pair row */
void vm_binop_apply(int32_t op, int32_t is_bool, int32_t drop_nan_right,
                    const double* a, const double* b, int64_t n,
                    int32_t has_fill_left, double fill_left,
                    int32_t has_fill_right, double fill_right,
                    double* out) {
  for (int64_t j = 0; j < n; j++) {
    double x = a[j], y = b[j];
    int ln = isnan(x), rn = isnan(y);
    if (ln && rn) {
      out[j] = vm_binop_scalar(op, is_bool, x, y);
      continue;
    }
    if (drop_nan_right && rn && !has_fill_right) {
      out[j] = bnan();
      continue;
    }
    if (ln && has_fill_left) x = fill_left;
    if (rn && has_fill_right) y = fill_right;
    out[j] = vm_binop_scalar(op, is_bool, x, y);
  }
}
