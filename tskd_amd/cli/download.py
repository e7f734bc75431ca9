"""PhysioNet waveform downloader — reference bin/download.py: batch
`wget -r` of mimic3wdb-matched per-patient directories. This environment has
no network; by default the commands are written to a shell script
(--execute actually runs them)."""

from __future__ import annotations

import argparse
import subprocess

BASE = "https://physionet.org/files/mimic3wdb-matched/1.0"


def build_commands(patients, out_dir: str):
    cmds = []
    for pid in patients:
        url = f"{BASE}/{pid[0:3]}/{pid}/"
        cmds.append(["wget", "-r", "-N", "-c", "-np", "-P", out_dir, url])
    return cmds


def main(argv=None) -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--patients", nargs="+", required=True,
                    help="patient ids, e.g. p000194 p044083")
    ap.add_argument("--out", default="data/waveform")
    ap.add_argument("--script", default="download_waveforms.sh")
    ap.add_argument("--execute", action="store_true")
    args = ap.parse_args(argv)
    cmds = build_commands(args.patients, args.out)
    if args.execute:
        for c in cmds:
            subprocess.run(c, check=True)
    else:
        with open(args.script, "w") as f:
            f.write("#!/bin/sh\n")
            for c in cmds:
                f.write(" ".join(c) + "\n")
        print(f"wrote {len(cmds)} wget commands to {args.script}")


if __name__ == "__main__":
    main()
