"""ctypes harness for tests: re-exports the product client
(kubebrain_amd.client) and adds the TEST-ONLY oracle loader.

The oracle (oracle/liboracle.so) is TEST INFRASTRUCTURE ONLY — it is loaded
here (tests), by __graft_entry__.smoke() and by bench.py's cpu_baseline leg,
never by the product package.
"""
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubebrain_amd.client import (  # noqa: F401
    BADKEY, CAS_FAILED, COMPACTED, ENOBUF, ENOGPU, INTERNAL, INVALID_ARG,
    KEYTOOLONG, NOTFOUND, OK, REV_DRIFT, UNCERTAIN, UNSUPPORTED, WATCH_DROPPED,
    WATCH_EMPTY, WATCH_LOW, Ev, Kv, RangeResp, Store, WriteResp, _parse_events,
    _parse_kvs)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def build_oracle() -> str:
    so = os.path.join(REPO, "oracle", "liboracle.so")
    src = [os.path.join(REPO, "oracle", f)
           for f in ("oracle.cc", "oracle_cabi.cc", "oracle.h")]
    if not os.path.exists(so) or any(os.path.getmtime(s) > os.path.getmtime(so)
                                     for s in src):
        subprocess.check_call(["make", "-C", os.path.join(REPO, "oracle")])
    return so


def open_oracle(**kw) -> Store:
    return Store(build_oracle(), "okb_", **kw)
