"""device-scheduler binary (reference cmd/device-scheduler/main.go):
lease leader election, scheduler-extender HTTP app, graceful stop.

Replicas race for the `vgpu-scheduler` Lease; every replica serves
HTTP (kube-scheduler load-balances to the Service), but non-leaders
answer mutating verbs (filter/bind/preempt) with 503 so only one
writer patches pod annotations — the reference achieves the same by
gating `runApp` on the election (main.go:256-271).
"""
from __future__ import annotations

import argparse
import logging
import os
import sys

from ..client.kube import RestKubeClient
from ..client.lease import LeaderElector
from ..scheduler.http import create_app
from ..util import consts


def main(argv=None):
    ap = argparse.ArgumentParser("vgpu-device-scheduler")
    ap.add_argument("--bind", default="0.0.0.0:3456")
    ap.add_argument("--apiserver", default=None)
    ap.add_argument("--domain", default=consts.AMD_DOMAIN)
    ap.add_argument("--leader-elect", action="store_true", default=False)
    ap.add_argument("--leader-elect-namespace",
                    default=os.environ.get("POD_NAMESPACE", "kube-system"))
    ap.add_argument("--leader-elect-name", default="vgpu-scheduler")
    args = ap.parse_args(argv)

    logging.basicConfig(level=logging.INFO)
    consts.set_domain(args.domain)
    client = RestKubeClient(base_url=args.apiserver)
    app = create_app(client)

    elector = None
    if args.leader_elect:
        elector = LeaderElector(
            client, args.leader_elect_namespace, args.leader_elect_name,
            identity=os.environ.get("POD_NAME") or None)
        elector.run_background()

        from fastapi import Request, Response

        @app.middleware("http")
        async def leader_gate(request: Request, call_next):
            mutating = request.url.path.startswith("/scheduler/") and \
                not request.url.path.endswith("filter-dryrun")
            if mutating and not elector.leading:
                return Response("not leader", status_code=503)
            return await call_next(request)

    import uvicorn
    host, port = args.bind.rsplit(":", 1)
    uvicorn.run(app, host=host, port=int(port))
    if elector:
        elector.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
