/* Host-compiles the PRODUCT's general script interpreter
 * (kv_script_host.inc over kv_validate_host.inc) for differential testing
 * against the oracle — test infrastructure only. */
#include <stdint.h>
#include <string.h>

#include "kaspa_engine_abi.h"
#include "kv_validate_host.inc"
#include "kv_script_host.inc"

extern "C" int host_run_input_script(const uint8_t *blob, size_t blob_len,
                                     uint32_t tx_index, uint32_t input_index,
                                     uint64_t mass_per_sig_op) {
  std::vector<kvhost::HTx> txs;
  int n = kvhost::parse_blob_host(blob, blob_len, txs);
  if (n < 0 || tx_index >= (uint32_t)n) return -1;
  if (input_index >= txs[tx_index].inputs.size()) return -1;
  return kvhost::kvh_run_input_script(txs[tx_index], txs[tx_index].inputs[input_index],
                                      input_index,
                                      mass_per_sig_op * 100ull);
}

/* seq-commit accessor mock (mirrors the oracle's harness shape: one known
 * chain block mapping to one commitment) */
static uint8_t g_seqc_block[32], g_seqc_commit[32];
static int g_seqc_set = 0;

extern "C" void host_set_seq_commit_mock(const uint8_t *block32,
                                         const uint8_t *commit32) {
  if (!block32) {
    g_seqc_set = 0;
    return;
  }
  memcpy(g_seqc_block, block32, 32);
  memcpy(g_seqc_commit, commit32, 32);
  g_seqc_set = 1;
}

static int seqc_mock_fn(void *, const uint8_t *block, uint8_t *commit) {
  if (memcmp(block, g_seqc_block, 32) != 0) return 1;
  memcpy(commit, g_seqc_commit, 32);
  return 0;
}

/* Collect/replay driver for the differential tests. memo = n_memo status
 * vectors (flat bytes + per-site sizes). On KVH_SCRIPT_SUSPEND the pending
 * requests are serialized 136B each:
 *   [0] ecdsa, [1] literal, [2] hash_type, [3] pad,
 *   [4..68) sig64, [68..101) pk33, [101..133) msg32, [133..136) pad.
 * Returns the script code, or -101 (suspend, requests written). */
extern "C" int host_run_input_script_collect(
    const uint8_t *blob, size_t blob_len, uint32_t tx_index, uint32_t input_index,
    uint64_t mass_per_sig_op, const uint8_t *memo_flat, const uint32_t *memo_sizes,
    uint32_t n_memo, uint8_t *pending_out, uint32_t pending_cap,
    uint32_t *n_pending) {
  std::vector<kvhost::HTx> txs;
  int n = kvhost::parse_blob_host(blob, blob_len, txs);
  if (n < 0 || tx_index >= (uint32_t)n) return -1;
  if (input_index >= txs[tx_index].inputs.size()) return -1;
  std::vector<std::vector<uint8_t>> memo(n_memo);
  size_t off = 0;
  for (uint32_t k = 0; k < n_memo; k++) {
    memo[k].assign(memo_flat + off, memo_flat + off + memo_sizes[k]);
    off += memo_sizes[k];
  }
  std::vector<kvhost::KvSigReq> pending;
  kvhost::KvsRunCtx rctx;
  rctx.memo = &memo;
  rctx.pending = &pending;
  rctx.seqc_fn = g_seqc_set ? seqc_mock_fn : nullptr;
  rctx.seqc_user = nullptr;
  int rc = kvhost::kvh_run_input_script(txs[tx_index],
                                        txs[tx_index].inputs[input_index],
                                        input_index, mass_per_sig_op * 100ull,
                                        &rctx);
  if (rc == kvhost::KVH_SCRIPT_SUSPEND) {
    uint32_t np = (uint32_t)pending.size();
    if (np > pending_cap) return -2;
    for (uint32_t r = 0; r < np; r++) {
      uint8_t *p = pending_out + (size_t)r * 136;
      memset(p, 0, 136);
      p[0] = pending[r].ecdsa;
      p[1] = pending[r].literal;
      p[2] = pending[r].hash_type;
      memcpy(p + 4, pending[r].sig, 64);
      memcpy(p + 68, pending[r].pk, 33);
      memcpy(p + 101, pending[r].msg, 32);
    }
    *n_pending = np;
  }
  return rc;
}
