"""device-webhook binary (reference cmd/device-webhook)."""
from __future__ import annotations

import argparse
import sys


def main(argv=None):
    ap = argparse.ArgumentParser("vgpu-device-webhook")
    ap.add_argument("--bind", default="0.0.0.0:8443")
    ap.add_argument("--tls-cert", default=None)
    ap.add_argument("--tls-key", default=None)
    ap.add_argument("--dra-mode", action="store_true")
    ap.add_argument("--dra-per-container", action="store_true",
                    help="emit one ResourceClaim per container instead "
                         "of one combined pod claim")
    args = ap.parse_args(argv)

    import uvicorn
    from ..webhook.admission import create_app

    client = None
    if args.dra_mode:
        # DRA conversion creates ResourceClaimTemplates server-side
        from ..client.kube import RestKubeClient
        client = RestKubeClient()
    host, port = args.bind.rsplit(":", 1)
    kwargs = {}
    if args.tls_cert and args.tls_key:
        kwargs = dict(ssl_certfile=args.tls_cert,
                      ssl_keyfile=args.tls_key)
    uvicorn.run(create_app(dra_mode=args.dra_mode, client=client,
                           dra_per_container=args.dra_per_container),
                host=host, port=int(port), **kwargs)
    return 0


if __name__ == "__main__":
    sys.exit(main())
