/* cubefs_amd/csrc/gfrs_gf.cpp — product-side GF(2^8) field + matrix math.
 *
 * Follows the reference algorithm (klauspost/reedsolomon v1.11.7 as
 * vendored by cubefs/cubefs):
 *   galois.go:13-26 field (generating polynomial 29 → 0x11D)
 *   galois.go:855-906 galMultiply/galDivide/galExp
 *   matrix.go:193-266 Gauss-Jordan inversion (pivot order preserved so the
 *     encode matrix is bit-identical)
 *   matrix.go:271-282 vandermonde
 *   reedsolomon.go:220-244 buildMatrix (the default; CubeFS never selects
 *     Cauchy/Jerasure/PAR1, reedsolomon.go:471-472)
 */
#include "gfrs_internal.h"

#include <cstring>

namespace gfrs {

GfTables::GfTables() {
  std::memset(log_t, 0, sizeof(log_t));
  int e = 1;
  for (int i = 0; i < 255; i++) {
    exp_t[i] = static_cast<uint8_t>(e);
    log_t[e] = static_cast<uint8_t>(i);
    e <<= 1;
    if (e & 0x100) e ^= 0x11D;
  }
  for (int i = 0; i < 255; i++) exp_t[255 + i] = exp_t[i];
  for (int a = 0; a < 256; a++)
    for (int b = 0; b < 256; b++)
      mul[a][b] =
          (a && b) ? exp_t[int(log_t[a]) + int(log_t[b])] : uint8_t(0);
  for (int c = 0; c < 256; c++)
    for (int x = 0; x < 16; x++) {
      lo[c][x] = mul[c][x];
      hi[c][x] = mul[c][x << 4];
    }
}

uint8_t GfTables::gexp(uint8_t a, int n) const {
  if (n == 0) return 1;
  if (a == 0) return 0;
  long r = long(log_t[a]) * n;
  while (r >= 255) r -= 255;
  return exp_t[r];
}

uint8_t GfTables::div(uint8_t a, uint8_t b) const {
  if (a == 0) return 0;
  int r = int(log_t[a]) - int(log_t[b]);
  if (r < 0) r += 255;
  return exp_t[r];
}

const GfTables &gft() {
  static GfTables t;
  return t;
}

bool gf_invert(const uint8_t *in, int n, uint8_t *out) {
  const GfTables &t = gft();
  const int cols = 2 * n;
  std::vector<uint8_t> w(size_t(n) * cols, 0);
  for (int r = 0; r < n; r++) {
    std::memcpy(&w[size_t(r) * cols], in + size_t(r) * n, n);
    w[size_t(r) * cols + n + r] = 1;
  }
  auto row = [&](int r) { return &w[size_t(r) * cols]; };
  for (int r = 0; r < n; r++) {
    if (row(r)[r] == 0) {
      for (int rb = r + 1; rb < n; rb++)
        if (row(rb)[r] != 0) {
          for (int c = 0; c < cols; c++) std::swap(row(r)[c], row(rb)[c]);
          break;
        }
    }
    if (row(r)[r] == 0) return false;
    if (row(r)[r] != 1) {
      uint8_t s = t.div(1, row(r)[r]);
      for (int c = 0; c < cols; c++) row(r)[c] = t.mul[row(r)[c]][s];
    }
    for (int rb = r + 1; rb < n; rb++) {
      uint8_t s = row(rb)[r];
      if (s)
        for (int c = 0; c < cols; c++) row(rb)[c] ^= t.mul[s][row(r)[c]];
    }
  }
  for (int d = 0; d < n; d++)
    for (int ra = 0; ra < d; ra++) {
      uint8_t s = row(ra)[d];
      if (s)
        for (int c = 0; c < cols; c++) row(ra)[c] ^= t.mul[s][row(d)[c]];
    }
  for (int r = 0; r < n; r++) std::memcpy(out + size_t(r) * n, row(r) + n, n);
  return true;
}

bool gf_build_matrix(int k, int total, uint8_t *out) {
  if (k <= 0 || total < k || total > 256) return false;
  const GfTables &t = gft();
  std::vector<uint8_t> vm(size_t(total) * k);
  for (int r = 0; r < total; r++)
    for (int c = 0; c < k; c++) vm[size_t(r) * k + c] = t.gexp(uint8_t(r), c);
  std::vector<uint8_t> inv(size_t(k) * k);
  if (!gf_invert(vm.data(), k, inv.data())) return false;
  for (int r = 0; r < total; r++)
    for (int c = 0; c < k; c++) {
      uint8_t v = 0;
      for (int i = 0; i < k; i++)
        v ^= t.mul[vm[size_t(r) * k + i]][inv[size_t(i) * k + c]];
      out[size_t(r) * k + c] = v;
    }
  return true;
}

}  // namespace gfrs
