import ctypes
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X (run via gpurun)")


def _build_oracle():
    subprocess.run(["make", "-s", "-C", os.path.join(REPO, "oracle")], check=True)


_oracle = None


def load_oracle():
    """The oracle shared library (test infrastructure; see oracle/oracle.h)."""
    global _oracle
    if _oracle is None:
        _build_oracle()
        _oracle = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    return _oracle


@pytest.fixture(scope="session")
def oracle():
    return load_oracle()


@pytest.fixture(scope="session")
def golden():
    def load(name):
        with open(os.path.join(REPO, "tests", "golden", name)) as f:
            return json.load(f)

    return load


def oracle_tx_id(lib, blob, index):
    out = (ctypes.c_uint8 * 32)()
    rc = lib.ok_tx_id(bytes(blob), len(blob), index, out)
    assert rc == 0
    return bytes(out)


def oracle_sighash(lib, blob, tx_index, input_index, hash_type, ecdsa=False):
    out = (ctypes.c_uint8 * 32)()
    rc = lib.ok_sighash(bytes(blob), len(blob), tx_index, input_index, hash_type,
                        1 if ecdsa else 0, out)
    assert rc == 0
    return bytes(out)
