"""Heartbeat supervisor tests: dead/stale processes are restarted
(the reference's dead code — main.py:417-473 — implemented for real)."""
import time

import torch.multiprocessing as mp

import main as main_mod


def _die_fast():
    raise SystemExit(1)


def _stamp_and_sleep(hb, stop_event):
    while not stop_event.is_set():
        hb.value = time.time()
        time.sleep(0.05)


def _silent(stop_event):
    while not stop_event.is_set():
        time.sleep(0.1)


def test_supervisor_restarts_dead_process():
    ctx = mp.get_context("spawn")
    stop = ctx.Event()
    sup = main_mod.Supervisor(stop)
    sup.spawn("dier", _die_fast, ())
    first = sup.specs["dier"]["proc"]
    first.join(10)
    assert not first.is_alive()
    # one monitor poll should respawn it
    import threading

    t = threading.Thread(target=sup.monitor, kwargs={"poll_s": 0.2}, daemon=True)
    t.start()
    deadline = time.time() + 40
    while time.time() < deadline:
        if sup.specs["dier"]["proc"] is not first:
            break
        time.sleep(0.1)
    assert sup.specs["dier"]["proc"] is not first, "dead process was not respawned"
    stop.set()
    sup.shutdown()


def test_supervisor_keeps_healthy_process():
    ctx = mp.get_context("spawn")
    stop = ctx.Event()
    hb = ctx.Value("d", time.time())
    sup = main_mod.Supervisor(stop)
    sup.spawn("healthy", _stamp_and_sleep, (hb, stop), heartbeat=hb)
    proc = sup.specs["healthy"]["proc"]
    import threading

    t = threading.Thread(target=sup.monitor, kwargs={"poll_s": 0.2}, daemon=True)
    t.start()
    time.sleep(2.0)
    assert sup.specs["healthy"]["proc"] is proc  # never restarted
    assert proc.is_alive()
    stop.set()
    sup.shutdown()


def test_supervisor_restarts_stale_heartbeat(monkeypatch):
    monkeypatch.setattr(main_mod, "HEARTBEAT_TIMEOUT_S", 0.5)
    ctx = mp.get_context("spawn")
    stop = ctx.Event()
    hb = ctx.Value("d", time.time() - 100.0)  # already stale
    sup = main_mod.Supervisor(stop)
    sup.spawn("stale", _silent, (stop,), heartbeat=hb)
    first = sup.specs["stale"]["proc"]
    import threading

    t = threading.Thread(target=sup.monitor, kwargs={"poll_s": 0.2}, daemon=True)
    t.start()
    deadline = time.time() + 40
    while time.time() < deadline:
        if sup.specs["stale"]["proc"] is not first:
            break
        time.sleep(0.1)
    assert sup.specs["stale"]["proc"] is not first, "stale process was not restarted"
    stop.set()
    sup.shutdown()
