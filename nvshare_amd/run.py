"""nvshare-run: launch any command as an nvshare-amd client.

    python -m nvshare_amd.run [options] -- <command> [args...]

Sets LD_PRELOAD=libnvshare.so, HSA_XNACK=1 and the scheduler socket
directory, then execs the command.  The bare-metal equivalent of what
the K8s device plugin injects into pods.
"""

from __future__ import annotations

import argparse
import os
import sys

from nvshare_amd.env import client_env


def main(argv: list[str] | None = None) -> None:
    argv = list(sys.argv[1:] if argv is None else argv)
    if "--" in argv:
        split = argv.index("--")
        opts, cmd = argv[:split], argv[split + 1:]
    else:
        opts, cmd = [], argv

    ap = argparse.ArgumentParser(prog="nvshare-run")
    ap.add_argument("--sock-dir", default=None)
    ap.add_argument("--debug", action="store_true")
    ap.add_argument("--oversubscribe", action="store_true")
    ap.add_argument("--standalone", action="store_true")
    ap.add_argument("--reserve-mib", type=int, default=None)
    ap.add_argument("--fake-total-mib", type=int, default=None)
    ap.add_argument("--prefetch", action="store_true")
    args = ap.parse_args(opts)

    if not cmd:
        ap.error("no command given (usage: nvshare-run [opts] -- cmd ...)")

    env = client_env(
        sock_dir=args.sock_dir,
        debug=args.debug,
        oversubscribe=args.oversubscribe,
        standalone=args.standalone,
        reserve_mib=args.reserve_mib,
        fake_total_mib=args.fake_total_mib,
        prefetch=True if args.prefetch else None,
    )
    os.execvpe(cmd[0], cmd, env)


if __name__ == "__main__":
    main()
