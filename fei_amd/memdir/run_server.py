"""memdir server launcher (reference parity: memdir_tools/run_server.py:24-57)."""

from __future__ import annotations

import argparse
import secrets


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(description="Run the memdir HTTP server")
    parser.add_argument("--port", type=int, default=5000)
    parser.add_argument("--host", default="127.0.0.1")
    parser.add_argument("--base", default=None, help="Memdir base directory")
    parser.add_argument("--api-key", default=None)
    parser.add_argument("--generate-key", action="store_true",
                        help="Print a fresh API key and exit")
    args = parser.parse_args(argv)

    if args.generate_key:
        print(secrets.token_hex(16))
        return 0

    from fei_amd.memdir.server import create_app

    app = create_app(base=args.base, api_key=args.api_key)
    app.run(host=args.host, port=args.port, threaded=True)
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
