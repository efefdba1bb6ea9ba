"""Extract the reference's canonical script vectors
(crypto/txscript/test-data/script_tests.json — run by the reference's own
test_bitcoind_tests at lib.rs:2620) plus the opcode name→byte table parsed
from the opcode_list! declarations, into tests/golden/. Data only."""
import json
import re
import shutil

SRC = '/root/reference/crypto/txscript/src/opcodes/mod.rs'
shutil.copy('/root/reference/crypto/txscript/test-data/script_tests.json',
            '/root/repo/tests/golden/script_tests.json')

text = open(SRC).read()
table = {}
aliases = {}
for m in re.finditer(r'opcode\s*(?:\|(\w+)\|\s*)?(\w+)<(0x[0-9a-fA-F]+)\s*,', text):
    alias, name, code = m.group(1), m.group(2), int(m.group(3), 16)
    table[name] = code
    if alias:
        aliases[alias] = code
print(f"{len(table)} opcodes, {len(aliases)} aliases")
assert table.get('OpFalse') == 0 and table.get('OpDup') is not None
json.dump({"opcodes": table, "aliases": aliases},
          open('/root/repo/tests/golden/script_opcodes.json', 'w'), indent=0)
print("written")
