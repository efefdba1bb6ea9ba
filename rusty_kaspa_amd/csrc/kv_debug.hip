/* Temporary hang-isolation kernel: writes progress markers into host-visible
 * pinned memory at each stage of one schnorr verification. */
#include "kv_secp_kernels.hip"
using namespace kv;

extern "C" __global__ void kv_debug_verify(volatile int *progress,
                                           const uint8_t *tuple, int *out) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  progress[0] = 1;
  /* 1. fe_mul loop */
  fe a = {{0x123456789abcdefULL, 0x2, 0x3, 0x4}};
  fe b = {{0xdeadbeefULL, 0x5, 0x6, 0x7}};
  fe r = a;
  for (int i = 0; i < 1000; i++) fe_mul(r, r, b);
  out[0] = (int)r.n[0];
  progress[0] = 2;
  /* 2. fe_inv */
  fe inv;
  fe_inv(inv, r);
  out[1] = (int)inv.n[0];
  progress[0] = 3;
  /* 3. sqrt */
  fe s;
  int has = fe_sqrt(s, r);
  out[2] = has;
  progress[0] = 4;
  /* 4. lift_x from tuple pk */
  ge P;
  int okp = lift_x_even(P, tuple + 64);
  out[3] = okp;
  progress[0] = 5;
  /* 5. sha challenge */
  uint8_t eh[32];
  sha256_tagged96(BIP340_CHALLENGE_MID, tuple, tuple + 64, tuple + 96, eh);
  out[4] = eh[0];
  progress[0] = 6;
  /* 6. scalar ops */
  sc ss, e, ne;
  sc_from_be(ss, tuple + 32);
  sc_from_be(e, eh);
  sc_neg(ne, e);
  progress[0] = 7;
  /* 7. 16 iterations of the ecmult inner loop shape */
  gej R;
  gej_set_infinity(R);
  for (int i = 0; i < 16; i++) {
    gej t;
    gej_double(t, R);
    R = t;
    gej_add_ge(t, R, P);
    R = t;
  }
  out[5] = (int)R.x.n[0];
  progress[0] = 8;
  /* 8. full ecmult */
  ecmult_double(R, ss, ne, P, progress);
  out[6] = (int)R.x.n[0];
  progress[0] = 9;
  /* 9. full verify */
  uint8_t st = schnorr_verify_one(tuple, tuple + 32, tuple + 64, tuple + 96);
  out[7] = st;
  progress[0] = 10;
}

extern "C" __global__ void kv_debug_pure(volatile int *progress, int iters, int *out) {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  /* fully synthetic: scalars from constants, no memory in the loop */
  sc ss = {{0x123456789abcdefULL, 0xfedcba9876543210ULL, 0x1111111122222222ULL,
            0x0123456701234567ULL}};
  sc ee = {{0xaaaabbbbccccddddULL, 0x1234123412341234ULL, 0x5678567856785678ULL,
            0x0feeddccbbaa0099ULL}};
  ge P = GE_G;
  gej R;
  gej_set_infinity(R);
  progress[0] = 100;
  for (int w = 3; w >= 0; w--) {
    u64 gw = ss.d[w], pw = ee.d[w];
    for (int b = 63; b >= 0; b--) {
      gej t;
      gej_double(t, R);
      R = t;
      gej_add_ge(t, R, GE_G);
      gej_cmov(R, t, (gw >> b) & 1);
      gej_add_ge(t, R, P);
      gej_cmov(R, t, (pw >> b) & 1);
    }
    progress[0] = 100 + (3 - w) + 1;
    if ((3 - w) * 64 >= iters) break;
  }
  out[0] = (int)R.x.n[0];
  progress[0] = 110;
}
