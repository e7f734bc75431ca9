"""Signal-availability survey — the reference's explore_fantasia.ipynb /
signal-survey workflow as a CLI.

Scans WFDB numerics records under a waveform tree (the reference's
`WAVEFPATH` layout: pXX/pXXXXXX/<record>.hea) and tabulates, per channel
name, how many records carry it and the per-record sample counts — the
"which channels can we actually train on" question the reference answered
with a notebook (explore_fantasia.ipynb cells 1, 10).

Usage:
    python -m tskd_amd.cli.surveysignals [--wavef-path DIR] [--json]
    python -m tskd_amd.cli.surveysignals --records p000194-2112-05-23-14-34n
"""

from __future__ import annotations

import argparse
import json
import os
import sys
from collections import Counter, defaultdict
from typing import Dict, List, Tuple

from tskd_amd.config import get_global_config
from tskd_amd.io import rdrecord


def find_numerics_records(wavef_path: str) -> List[str]:
    """All numerics record paths (header basename ends with 'n', the
    MIMIC numerics convention) under the pXX/pXXXXXX tree."""
    out = []
    for root, _dirs, files in os.walk(wavef_path):
        for f in sorted(files):
            if f.endswith("n.hea"):
                out.append(os.path.join(root, f[: -len(".hea")]))
    return sorted(out)


def survey(record_paths: List[str]) -> Tuple[Counter, Dict[str, dict], list]:
    counts: Counter = Counter()
    per_record: Dict[str, dict] = {}
    errors = []
    for rp in record_paths:
        try:
            rec = rdrecord(rp)
        except Exception as e:  # unreadable/partial record: report, move on
            errors.append((os.path.basename(rp), str(e)))
            continue
        import numpy as np
        present = {}
        for i, name in enumerate(rec.sig_name):
            col = rec.p_signal[:, i]
            n_valid = int(np.isfinite(col).sum())
            present[name] = n_valid
            if n_valid:
                counts[name] += 1
        per_record[os.path.basename(rp)] = {
            "fs_hz": rec.fs, "sig_len": rec.sig_len,
            "channels": present,
        }
    return counts, per_record, errors


def main(argv=None) -> None:
    cfg = get_global_config()
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--wavef-path", default=cfg.wavef_path)
    ap.add_argument("--records", nargs="*", default=None,
                    help="explicit record names instead of scanning")
    ap.add_argument("--json", action="store_true")
    args = ap.parse_args(argv)

    if args.records:
        from tskd_amd.io import get_waveform_path
        paths = [r if os.path.sep in r else get_waveform_path(r, cfg)
                 for r in args.records]
    else:
        paths = find_numerics_records(args.wavef_path)
    if not paths:
        print(f"no numerics records under {args.wavef_path}",
              file=sys.stderr)
        sys.exit(1)
    counts, per_record, errors = survey(paths)
    configured = set(cfg.channel_names)
    if args.json:
        print(json.dumps({"n_records": len(per_record),
                          "channel_counts": dict(counts),
                          "records": per_record,
                          "errors": errors}, indent=1))
        return
    print(f"{len(per_record)} numerics record(s) under {args.wavef_path}")
    print(f"{'channel':24s} {'records':>8s}  {'configured':>10s}")
    seen = set()
    for name, n in counts.most_common():
        seen.add(name)
        mark = "yes" if name in configured else "-"
        print(f"{name:24s} {n:8d}  {mark:>10s}")
    for name in sorted(configured - seen):
        print(f"{name:24s} {0:8d}  {'yes':>10s}   (configured, absent)")
    for rp, err in errors:
        print(f"! {rp}: {err}", file=sys.stderr)


if __name__ == "__main__":
    main()
