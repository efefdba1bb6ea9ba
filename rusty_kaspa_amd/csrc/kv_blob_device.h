/* Device-side view of the transaction-batch blob (format:
 * include/kaspa_engine_abi.h). The blob is uploaded to HBM verbatim; kernels
 * parse lazily through these cursors — no host re-layout, the format is
 * already flat and little-endian. PRODUCT code. */
#ifndef KV_BLOB_DEVICE_H
#define KV_BLOB_DEVICE_H

#ifndef KV_HOST_TEST
#include <hip/hip_runtime.h>
#endif
#include <stdint.h>

namespace kv {

__device__ __forceinline__ uint16_t bl_rd16(const uint8_t *p) {
  return (uint16_t)p[0] | ((uint16_t)p[1] << 8);
}
__device__ __forceinline__ uint32_t bl_rd32(const uint8_t *p) {
  return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
         ((uint32_t)p[3] << 24);
}
__device__ __forceinline__ uint64_t bl_rd64(const uint8_t *p) {
  uint64_t v = 0;
#pragma unroll
  for (int i = 0; i < 8; i++) v |= (uint64_t)p[i] << (8 * i);
  return v;
}

struct blob_tx {
  const uint8_t *base;      /* tx start */
  uint16_t version;
  uint16_t n_inputs;
  uint16_t n_outputs;
  uint64_t lock_time;
  const uint8_t *subnetwork_id; /* 20 */
  uint32_t payload_len;
  uint64_t gas;
  const uint8_t *tx_id;     /* 32 */
  const uint8_t *payload;
  const uint8_t *inputs0;   /* first input record */
};

struct blob_input {
  const uint8_t *prev_tx_id; /* 32 */
  uint32_t prev_index;
  uint64_t sequence;
  uint8_t commit_kind;
  uint16_t commit_value;
  const uint8_t *sig_script;
  uint32_t sig_script_len;
  uint64_t utxo_amount;
  uint64_t utxo_daa_score;
  uint8_t utxo_is_coinbase;
  uint8_t utxo_has_cov;
  uint16_t utxo_spk_version;
  const uint8_t *utxo_spk;
  uint32_t utxo_spk_len;
  const uint8_t *utxo_cov_id;
  const uint8_t *end; /* next record */
};

struct blob_output {
  uint64_t value;
  uint16_t spk_version;
  const uint8_t *spk;
  uint32_t spk_len;
  uint8_t has_covenant;
  uint16_t cov_auth_input;
  const uint8_t *cov_id;
  const uint8_t *end;
};

__device__ __forceinline__ void blob_tx_at(const uint8_t *blob, uint32_t tx_index,
                                           blob_tx &t) {
  uint32_t off = bl_rd32(blob + 4 + 4u * tx_index);
  const uint8_t *p = blob + off;
  t.base = p;
  t.version = bl_rd16(p);
  t.n_inputs = bl_rd16(p + 2);
  t.n_outputs = bl_rd16(p + 4);
  t.lock_time = bl_rd64(p + 8);
  t.subnetwork_id = p + 16;
  t.payload_len = bl_rd32(p + 36);
  t.gas = bl_rd64(p + 40);
  /* storage_mass at +48 */
  t.tx_id = p + 56;
  t.payload = p + 88;
  t.inputs0 = t.payload + t.payload_len;
}

__device__ __forceinline__ void blob_input_at(const uint8_t *p, blob_input &in) {
  in.prev_tx_id = p;
  in.prev_index = bl_rd32(p + 32);
  in.sequence = bl_rd64(p + 36);
  in.commit_kind = p[44];
  in.commit_value = bl_rd16(p + 46);
  in.sig_script_len = bl_rd32(p + 48);
  in.sig_script = p + 52;
  const uint8_t *q = in.sig_script + in.sig_script_len;
  in.utxo_amount = bl_rd64(q);
  in.utxo_daa_score = bl_rd64(q + 8);
  in.utxo_is_coinbase = q[16];
  in.utxo_has_cov = q[17];
  in.utxo_spk_version = bl_rd16(q + 18);
  in.utxo_spk_len = bl_rd32(q + 20);
  in.utxo_spk = q + 24;
  in.utxo_cov_id = in.utxo_spk + in.utxo_spk_len;
  in.end = in.utxo_cov_id + (in.utxo_has_cov ? 32 : 0);
}

__device__ __forceinline__ void blob_output_at(const uint8_t *p, blob_output &o) {
  o.value = bl_rd64(p);
  o.spk_version = bl_rd16(p + 8);
  o.spk_len = bl_rd32(p + 12);
  o.spk = p + 16;
  const uint8_t *q = o.spk + o.spk_len;
  o.has_covenant = q[0];
  if (o.has_covenant) {
    o.cov_auth_input = bl_rd16(q + 1);
    o.cov_id = q + 3;
    o.end = q + 35;
  } else {
    o.cov_auth_input = 0;
    o.cov_id = nullptr;
    o.end = q + 1;
  }
}

} // namespace kv

#endif
