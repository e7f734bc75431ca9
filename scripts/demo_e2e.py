#!/usr/bin/env python3
"""Multi-PROCESS end-to-end demo: the real CLI entrypoints as separate OS
processes sharing a bus directory (the docker-compose topology), under the
supervisor, against a synthetic WFDB record. Exits 0 when predictions land
in the store.

Usage: python scripts/demo_e2e.py [workdir]
"""

import os
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import numpy as np  # noqa: E402


def write_record(root, record, sig_names, nsamp=40):
    pid = record[0:7]
    d = os.path.join(root, pid[0:3], pid)
    os.makedirs(d, exist_ok=True)
    rng = np.random.default_rng(0)
    adc = rng.normal(700, 80, size=(nsamp, len(sig_names))).astype(np.int16)
    adc.tofile(os.path.join(d, f"{record}.dat"))
    with open(os.path.join(d, f"{record}.hea"), "w") as f:
        f.write(f"{record} {len(sig_names)} 0.0166666666667 {nsamp} "
                f"14:34:23.221 23/05/2112\n")
        for name in sig_names:
            f.write(f"{record}.dat 16 10/bpm 16 0 736 0 0 {name}\n")


def main() -> int:
    work = sys.argv[1] if len(sys.argv) > 1 else tempfile.mkdtemp(
        prefix="tskd_demo_")
    os.makedirs(work, exist_ok=True)
    bus = os.path.join(work, "bus")
    store = os.path.join(work, "predictions.log")
    wavef = os.path.join(work, "wavef")
    record = "p000194-demo"
    channels = ["HR", "RESP", "PULSE", "SpO2"]
    write_record(wavef, record, channels)

    cfg_path = os.path.join(work, "config.cfg")
    with open(cfg_path, "w") as f:
        f.write("[PATHS]\nMOUNTPATH = .\n"
                f"WAVEFPATH = {wavef}\nMODELPATH = missing.pth\n"
                "[SETTINGS]\nUSE_CUDA = 0\nNUM_WORKERS = 0\n"
                f"CHANNEL_NAMES = {', '.join(channels)}\n"
                f"PATIENTRECORDS = {record}\n"
                "WINDOWSIZE = 120\nRECORDOVERLAP = 0.4\nBATCHSIZE = 16\n")

    env = dict(os.environ, PYTHONPATH=REPO, TSKD_CONFIG=cfg_path)
    procs = []

    def spawn(mod, *args):
        p = subprocess.Popen([sys.executable, "-m", mod, *args], env=env,
                             cwd=work)
        procs.append(p)
        return p

    try:
        # stages under the supervisor (restart-on-failure + offset resume)
        spawn("tskd_amd.parallel.supervisor", "--stage", "processstream",
              "--", "--bus-dir", bus, "--starting", "earliest",
              "--offsets-file", os.path.join(work, "proc.off"),
              "--device", "cpu", "--speed", "1000", "--max-triggers", "200")
        spawn("tskd_amd.parallel.supervisor", "--stage", "predictstream",
              "--", "--bus-dir", bus, "--starting", "earliest",
              "--offsets-file", os.path.join(work, "pred.off"),
              "--store-path", store, "--device", "cpu", "--speed", "1000",
              "--max-triggers", "200")
        time.sleep(2.0)
        # producer (very high speed => no sleeps)
        prod = spawn("tskd_amd.cli.sendstream", "--bus-dir", bus,
                     "--speed", "1e6",
                     "--log-file", os.path.join(work, "producer.log"))
        prod.wait(120)

        from tskd_amd.store import PredictionStore
        deadline = time.time() + 90
        n = 0
        while time.time() < deadline:
            n = PredictionStore(store).count()
            if n > 0:
                break
            time.sleep(0.5)
        print(f"[demo] predictions in store: {n}")
        if n == 0:
            return 1
        t, risk = PredictionStore(store).latest("p000194")
        print(f"[demo] latest p000194 risk={risk:.4f} @ {t}")
        return 0
    finally:
        for p in procs:
            if p.poll() is None:
                p.terminate()
        for p in procs:
            try:
                p.wait(10)
            except subprocess.TimeoutExpired:
                p.kill()


if __name__ == "__main__":
    sys.exit(main())
