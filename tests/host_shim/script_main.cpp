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
