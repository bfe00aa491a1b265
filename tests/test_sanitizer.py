"""Sanitizer + soak tier (the reference runs no -race or
sanitizer anywhere; this framework adds one).

- ASAN: the C++ scheduler/prefix-cache extension is rebuilt with
  AddressSanitizer and driven through a preemption+prefix-cache-churn
  workload in a subprocess (LD_PRELOAD=libasan); any heap error aborts
  the subprocess and fails the test.
- Soak: the control plane is hammered from many threads mixing sync
  executes, async executes, status polls and memory ops; every request
  must terminate cleanly (no 5xx, no stuck executions).
"""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent

ASAN_DRIVER = r"""
import random, sys
sys.path.insert(0, {root!r})
# import the ASAN build under the name the adapter expects
import importlib.util
spec = importlib.util.spec_from_file_location("agentfield_amd._native",
                                              {so!r})
mod = importlib.util.module_from_spec(spec)
spec.loader.exec_module(mod)
sys.modules["agentfield_amd._native"] = mod

from agentfield_amd.engine.prefix_cache import prefix_hashes

rng = random.Random(0)
nat = mod.NativeScheduler(4, 64, 4, 24, 64, True)  # tiny pool, prefix on
alive = {{}}
next_id = 0
for step in range(4000):
    if rng.random() < 0.5 and len(alive) < 40:
        plen = rng.randint(4, 18)
        prompt = [rng.randrange(40) for _ in range(plen)]
        if rng.random() < 0.5 and alive:
            # heavy prefix repetition drives cache sharing + eviction
            prompt = ([7, 3, 9, 1, 8, 2, 6, 4] * 3)[:plen]
        if nat.add(next_id, plen, prefix_hashes(prompt, 4)):
            alive[next_id] = rng.randint(1, 12)
        next_id += 1
    r = nat.schedule()
    if r.has_work:
        for sid in list(r.seq_ids):
            nat.note_token(sid)
            alive[sid] -= 1
            if alive[sid] <= 0:
                nat.finish(sid)
                del alive[sid]
print("steps ok; preempted", nat.n_preempted(),
      "cache pages", nat.cache_pages())
"""


def test_native_scheduler_under_asan(tmp_path):
    libasan = subprocess.run(["gcc", "-print-file-name=libasan.so"],
                             capture_output=True, text=True).stdout.strip()
    if not libasan or not Path(libasan).exists():
        pytest.skip("libasan not available")
    from agentfield_amd.native_build import build_asan
    so = build_asan(verbose=False)
    driver = tmp_path / "driver.py"
    driver.write_text(ASAN_DRIVER.format(root=str(ROOT), so=str(so)))
    r = subprocess.run(
        [sys.executable, str(driver)], capture_output=True, text=True,
        env={**os.environ, "LD_PRELOAD": libasan,
             # torch isn't imported here; leak checking off keeps the
             # CPython-level noise out — we want heap ERRORS
             "ASAN_OPTIONS": "detect_leaks=0,abort_on_error=1"},
        timeout=240)
    assert r.returncode == 0, f"ASAN failure:\n{r.stdout}\n{r.stderr[-4000:]}"
    assert "steps ok" in r.stdout
    assert "preempted" in r.stdout


def test_controlplane_threaded_soak():
    """Mixed-operation soak from many threads against a live plane."""
    import threading
    import httpx
    sys.path.insert(0, str(ROOT / "tests"))
    from helpers import AppServer
    from agentfield_amd.controlplane import ControlPlane, create_app
    from agentfield_amd.controlplane.server import Config
    from agentfield_amd.sdk import Agent

    cp = ControlPlane(Config(background_services=False, did_enabled=False,
                             sync_timeout=30.0))
    srv = AppServer(create_app(cp)).start().wait_healthy()
    agent = Agent("soak", agentfield_url=srv.base_url, auto_register=False)

    @agent.reasoner()
    def echo(v: int):
        return {"v": v}

    a_srv = AppServer(agent).start()
    agent.base_url = a_srv.base_url
    assert agent.register()

    errors: list[str] = []
    pending_async: list[str] = []
    lock = threading.Lock()

    def worker(wid: int):
        with httpx.Client(timeout=35.0) as c:
            for i in range(25):
                kind = i % 4
                try:
                    if kind == 0:
                        r = c.post(srv.base_url +
                                   "/api/v1/execute/soak.echo",
                                   json={"input": {"v": i}})
                        ok = (r.status_code == 200 and
                              r.json()["status"] == "completed")
                    elif kind == 1:
                        r = c.post(srv.base_url +
                                   "/api/v1/execute/async/soak.echo",
                                   json={"input": {"v": i}})
                        ok = r.status_code in (202, 503)
                        if r.status_code == 202:
                            with lock:
                                pending_async.append(
                                    r.json()["execution_id"])
                    elif kind == 2:
                        r = c.post(srv.base_url + "/api/v1/memory/set",
                                   json={"key": f"k{wid}", "value": i,
                                         "scope": "global"})
                        ok = r.status_code == 200
                    else:
                        r = c.get(srv.base_url + "/api/ui/v1/executions",
                                  params={"limit": 5})
                        ok = r.status_code == 200
                    if not ok:
                        with lock:
                            errors.append(f"w{wid} i{i}: {r.status_code} "
                                          f"{r.text[:120]}")
                except Exception as e:  # noqa: BLE001
                    with lock:
                        errors.append(f"w{wid} i{i}: {type(e).__name__} {e}")

    threads = [threading.Thread(target=worker, args=(w,)) for w in range(12)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    assert not errors, errors[:8]
    # every accepted async execution reaches a terminal state
    import time
    deadline = time.time() + 60
    remaining = set(pending_async)
    with httpx.Client(timeout=10.0) as c:
        while remaining and time.time() < deadline:
            for eid in list(remaining):
                st = c.get(srv.base_url +
                           f"/api/v1/executions/{eid}").json()["status"]
                if st in ("completed", "failed", "timeout", "cancelled"):
                    assert st == "completed", f"{eid}: {st}"
                    remaining.discard(eid)
            time.sleep(0.2)
    assert not remaining, f"stuck async executions: {list(remaining)[:5]}"
    a_srv.stop()
    srv.stop()


TSAN_DRIVER = r"""
import sys, threading
sys.path.insert(0, {root!r})
import importlib.util
spec = importlib.util.spec_from_file_location("agentfield_amd._native",
                                              {so!r})
mod = importlib.util.module_from_spec(spec)
spec.loader.exec_module(mod)

import os
seed = os.urandom(32)
pub = mod.ed25519_pubkey(seed)
key = os.urandom(32)

def worker(i):
    # the crypto helpers run under FastAPI worker threads in production;
    # hammer them concurrently (GIL releases inside OpenSSL calls)
    for j in range(300):
        msg = (f"m{{i}}-{{j}}").encode() * 8
        sig = mod.ed25519_sign(seed, msg)
        assert mod.ed25519_verify(pub, msg, sig)
        ct = mod.aes_gcm_encrypt(key, msg)
        assert mod.aes_gcm_decrypt(key, ct) == msg

threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
for t in threads:
    t.start()
for t in threads:
    t.join()
print("TSAN_DRIVER_OK")
"""


def test_native_crypto_under_tsan(tmp_path):
    """ThreadSanitizer over the Ed25519/AES-GCM helpers hammered from 8
    threads.  CPython itself is not TSAN-annotated, so only reports
    whose stacks hit OUR extension fail the test."""
    libtsan = subprocess.run(["gcc", "-print-file-name=libtsan.so"],
                             capture_output=True, text=True).stdout.strip()
    if not libtsan or not Path(libtsan).exists():
        pytest.skip("libtsan not available")
    from agentfield_amd.native_build import build_tsan
    so = build_tsan(verbose=False)
    script = tmp_path / "drv.py"
    script.write_text(TSAN_DRIVER.format(root=str(ROOT), so=str(so)))
    log = tmp_path / "tsan.log"
    r = subprocess.run(
        [sys.executable, str(script)],
        env={**os.environ, "LD_PRELOAD": libtsan,
             "TSAN_OPTIONS": f"log_path={log} exitcode=0 halt_on_error=0"},
        capture_output=True, text=True, timeout=300)
    assert "TSAN_DRIVER_OK" in r.stdout, r.stderr[-2000:]
    reports = "".join(p.read_text() for p in tmp_path.glob("tsan.log.*"))
    ours = [blk for blk in reports.split("==================")
            if "_native_tsan" in blk and "data race" in blk.lower()]
    assert not ours, ours[0][:2000]
