#include "shim.h"
#include "kv_hash_device.h"
#include "kv_secp_device.h"
// pull in the verify-one functions without the kernels: define away kernel bodies
#define kv_schnorr_verify_kernel kv_schnorr_verify_kernel_unused
#define kv_ecdsa_verify_kernel kv_ecdsa_verify_kernel_unused
static unsigned long long __ballot(int) { return 0; }
struct dim3s { unsigned x, y, z; };
static dim3s blockIdx{0,0,0}, blockDim{1,1,1}, threadIdx{0,0,0};
#include "kv_secp_kernels.hip"
#include <stdio.h>
#include <string.h>
extern "C" {
int host_glv_split(const uint64_t k_le[4], uint64_t out[8]) {
  kv::sc s; memcpy(s.d, k_le, 32);
  kv::glv_half h1, h2;
  kv::glv_split(s, h1, h2);
  out[0]=h1.d[0]; out[1]=h1.d[1]; out[2]=h1.d[2]; out[3]=h1.neg;
  out[4]=h2.d[0]; out[5]=h2.d[1]; out[6]=h2.d[2]; out[7]=h2.neg;
  return 0;
}
int host_init_gtable() {
  /* emulate kv_ec_table_init_kernel on host */
  using namespace kv;
  ge G;
  {
    fe gx, gy;
    for (int i = 0; i < 4; i++) { gx.n[i] = GE_G_X[i]; gy.n[i] = GE_G_Y[i]; }
    fe26_from_fe(G.x, gx);
    fe26_from_fe(G.y, gy);
  }
  gej acc;
  acc.x = G.x; acc.y = G.y; fe26_set_int(acc.z, 1);
  KV_G_TABLE8[0] = G;
  static fe26 zs[256], pref[256];
  for (int k = 1; k <= 255; k++) {
    KV_G_TABLE8[k].x = acc.x;
    KV_G_TABLE8[k].y = acc.y;
    zs[k] = acc.z;
    gej t;
    gej_add_ge(t, acc, G);
    acc = t;
  }
  pref[1] = zs[1];
  for (int k = 2; k <= 255; k++) fe26_mul(pref[k], pref[k - 1], zs[k]);
  fe26 inv;
  fe26_inv(inv, pref[255]);
  for (int k = 255; k >= 1; k--) {
    fe26 zi;
    if (k > 1) { fe26_mul(zi, inv, pref[k - 1]); fe26_mul(inv, inv, zs[k]); }
    else zi = inv;
    fe26 zi2, zi3;
    fe26_sqr(zi2, zi);
    fe26_mul(zi3, zi2, zi);
    fe26_mul(KV_G_TABLE8[k].x, KV_G_TABLE8[k].x, zi2);
    fe26_mul(KV_G_TABLE8[k].y, KV_G_TABLE8[k].y, zi3);
    fe26_normalize(KV_G_TABLE8[k].x);
    fe26_normalize(KV_G_TABLE8[k].y);
  }
  return 1;
}
int host_schnorr_verify(const uint8_t* pk, const uint8_t* msg, const uint8_t* sig) {
  uint8_t st = kv::schnorr_verify_one(sig, sig+32, pk, msg);
  return st == 0 ? 1 : (st == 2 ? -1 : 0);
}
int host_ecdsa_verify(const uint8_t* pk33, const uint8_t* msg, const uint8_t* sig) {
  uint8_t st = kv::ecdsa_verify_one(sig, sig+32, pk33, msg);
  return st == 0 ? 1 : (st == 2 ? -1 : (st == 3 ? -2 : 0));
}
void host_fe_mul(const uint64_t* a, const uint64_t* b, uint64_t* r) {
  kv::fe fa, fb, fr;
  memcpy(fa.n, a, 32); memcpy(fb.n, b, 32);
  kv::fe_mul(fr, fa, fb);
  memcpy(r, fr.n, 32);
}
void host_sc_mul(const uint64_t* a, const uint64_t* b, uint64_t* r) {
  kv::sc fa, fb, fr;
  memcpy(fa.d, a, 32); memcpy(fb.d, b, 32);
  kv::sc_mul(fr, fa, fb);
  memcpy(r, fr.d, 32);
}
}
extern "C" void host_fe26_mul(const uint32_t a[10], const uint32_t b[10], uint32_t r[10]) {
  kv::fe26 fa, fb, fr;
  for (int i = 0; i < 10; i++) { fa.l[i] = a[i]; fb.l[i] = b[i]; }
  kv::fe26_mul(fr, fa, fb);
  for (int i = 0; i < 10; i++) r[i] = fr.l[i];
}
extern "C" void host_fe26_sqr(const uint32_t a[10], uint32_t r[10]) {
  kv::fe26 fa, fr;
  for (int i = 0; i < 10; i++) fa.l[i] = a[i];
  kv::fe26_sqr(fr, fa);
  for (int i = 0; i < 10; i++) r[i] = fr.l[i];
}
extern "C" void host_fe26_normalize(uint32_t a[10]) {
  kv::fe26 fa;
  for (int i = 0; i < 10; i++) fa.l[i] = a[i];
  kv::fe26_normalize(fa);
  for (int i = 0; i < 10; i++) a[i] = fa.l[i];
}
extern "C" void host_fe26_roundtrip(const uint64_t a[4], uint64_t r[4]) {
  kv::fe fa, fr; kv::fe26 t;
  for (int i = 0; i < 4; i++) fa.n[i] = a[i];
  kv::fe26_from_fe(t, fa);
  kv::fe26_normalize(t);
  kv::fe26_to_fe(fr, t);
  for (int i = 0; i < 4; i++) r[i] = fr.n[i];
}
extern "C" void host_fe26_neg_norm(const uint32_t a[10], uint32_t m, uint32_t r[10]) {
  kv::fe26 fa, fr;
  for (int i = 0; i < 10; i++) fa.l[i] = a[i];
  kv::fe26_neg(fr, fa, m);
  kv::fe26_normalize(fr);
  for (int i = 0; i < 10; i++) r[i] = fr.l[i];
}
extern "C" void host_fe26_inv(const uint32_t a[10], uint32_t r[10]) {
  kv::fe26 fa, fr;
  for (int i = 0; i < 10; i++) fa.l[i] = a[i];
  kv::fe26_inv(fr, fa);
  kv::fe26_normalize(fr);
  for (int i = 0; i < 10; i++) r[i] = fr.l[i];
}
extern "C" int host_fe26_sqrt(const uint32_t a[10], uint32_t r[10]) {
  kv::fe26 fa, fr;
  for (int i = 0; i < 10; i++) fa.l[i] = a[i];
  int ok = kv::fe26_sqrt(fr, fa);
  kv::fe26_normalize(fr);
  for (int i = 0; i < 10; i++) r[i] = fr.l[i];
  return ok;
}
