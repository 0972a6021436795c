"""Session construction from flags + profile file.

Role-parity: sliceconfig/sliceconfig.go (Parse: flags + $HOME/.bigslice
profile -> exec.Session, pprof/status HTTP handlers) and exec/config.go
(profile keys).  Profile file: $HOME/.bigslice_amd/config, simple
`key = value` lines; recognized keys: parallelism, device, trace-path,
distributed, groupby-initial-cap, device-chunk-rows.
"""

from __future__ import annotations

import argparse
import os
from typing import Dict, Optional, Tuple

from .runtime.session import Session, start


def _load_profile(path: Optional[str] = None) -> Dict[str, str]:
    if path is None:
        path = os.path.join(os.path.expanduser("~"), ".bigslice_amd",
                            "config")
    out: Dict[str, str] = {}
    if not os.path.exists(path):
        return out
    with open(path) as fp:
        for line in fp:
            line = line.strip()
            if not line or line.startswith("#"):
                continue
            if "=" in line:
                k, v = line.split("=", 1)
                out[k.strip()] = v.strip()
    return out


def parse(argv=None, profile_path: str = None,
          http_port: int = None) -> Tuple[Session, list]:
    """Build a Session from profile + command-line flags; returns
    (session, remaining_args).  Mirrors sliceconfig.Parse()."""
    prof = _load_profile(profile_path)
    ap = argparse.ArgumentParser(add_help=False)
    ap.add_argument("--parallelism", type=int,
                    default=int(prof.get("parallelism", 0)) or None)
    ap.add_argument("--device", type=str,
                    default=prof.get("device") or None)
    ap.add_argument("--trace-path", type=str,
                    default=prof.get("trace-path") or None)
    ap.add_argument("--local", action="store_true",
                    default=prof.get("distributed", "") not in
                    ("1", "true"))
    ap.add_argument("--http", type=int, default=http_port)
    args, rest = ap.parse_known_args(argv)

    distributed = None if args.local else True
    sess = start(parallelism=args.parallelism, device=args.device,
                 distributed=distributed, trace_path=args.trace_path)
    if args.http:
        from .utils.debug_http import serve_session
        serve_session(sess, args.http)
    return sess, rest
