import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

ORACLE_SO = os.path.join(REPO, "oracle", "liboracle.so")
REF_CRC_SO = os.path.join(REPO, "oracle", "_ref", "libcrc_ref.so")
HIP_SO = os.path.join(REPO, "incubator_pegasus_amd", "csrc", "librrdb_hip.so")


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (gfx950) GPU")


def _ensure_oracle():
    if not os.path.exists(ORACLE_SO):
        subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                       capture_output=True)
    return ORACLE_SO


@pytest.fixture(scope="session")
def oracle_lib():
    """The CPU oracle — parity checker only (never the product path)."""
    from incubator_pegasus_amd.capi import RrdbLib

    return RrdbLib(_ensure_oracle())


@pytest.fixture(scope="session")
def hip_lib():
    from incubator_pegasus_amd import hip_lib as _hip

    return _hip()


@pytest.fixture()
def oracle_part(oracle_lib):
    p = oracle_lib.open(1, 0, -1)
    yield p
    p.close()
