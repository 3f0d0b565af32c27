"""client_env() unit tests (no GPU)."""

from __future__ import annotations

from nvshare_amd.env import client_env


def test_defaults(artifacts):
    env = client_env(base={})
    assert env["HSA_XNACK"] == "1"
    assert env["LD_PRELOAD"].endswith("libnvshare.so")
    assert "NVSHARE_ENABLE_SINGLE_OVERSUB" not in env


def test_preload_prepends(artifacts):
    env = client_env(base={"LD_PRELOAD": "/lib/other.so"})
    parts = env["LD_PRELOAD"].split(":")
    assert parts[0].endswith("libnvshare.so")
    assert parts[1] == "/lib/other.so"


def test_knobs(artifacts):
    env = client_env(base={}, sock_dir="/tmp/x", debug=True,
                     oversubscribe=True, standalone=True,
                     reserve_mib=123, fake_total_mib=456, prefetch=False,
                     evict=True, disable_um=True)
    assert env["NVSHARE_SOCK_DIR"] == "/tmp/x"
    assert env["NVSHARE_DEBUG"] == "1"
    assert env["NVSHARE_ENABLE_SINGLE_OVERSUB"] == "1"
    assert env["NVSHARE_STANDALONE"] == "1"
    assert env["NVSHARE_RESERVE_MIB"] == "123"
    assert env["NVSHARE_FAKE_TOTAL_MIB"] == "456"
    assert env["NVSHARE_PREFETCH"] == "0"
    assert env["NVSHARE_EVICT"] == "1"
    assert env["NVSHARE_DISABLE_UM"] == "1"


def test_stub_library_path(artifacts):
    env = client_env(base={"LD_LIBRARY_PATH": "/usr/lib"}, use_stub=True)
    parts = env["LD_LIBRARY_PATH"].split(":")
    assert parts[0] == str(artifacts.stub_dir)
    assert parts[1] == "/usr/lib"
