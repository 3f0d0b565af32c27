#!/usr/bin/env python3
"""nvshare-top: live scheduler status (watch-style operator view).

    python tools/nvshare_top.py [--sock-dir DIR] [--interval S] [--once]

Polls the STATUS endpoint (the wire extension nvsharectl -q uses) and
renders scheduling state, client count, queue depth and the tracked
memory total.  Complements the SIGUSR1 stderr dump with a remote,
non-intrusive view.
"""

from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from nvshare_amd import ctl  # noqa: E402


def render(st) -> str:
    return (f"scheduling={'on' if st.scheduling_on else 'OFF'} "
            f"tq={st.tq_seconds}s clients={st.clients} "
            f"queued={st.queued} tracked={st.tracked_mib} MiB")


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--sock-dir", default=None)
    ap.add_argument("--interval", type=float, default=2.0)
    ap.add_argument("--once", action="store_true")
    args = ap.parse_args()

    while True:
        try:
            st = ctl.status(args.sock_dir, timeout=5)
            line = render(st)
        except Exception as e:  # noqa: BLE001 — operator tool keeps going
            line = f"(scheduler unreachable: {e})"
        print(time.strftime("%H:%M:%S"), line, flush=True)
        if args.once:
            return 0
        time.sleep(args.interval)


if __name__ == "__main__":
    sys.exit(main())
