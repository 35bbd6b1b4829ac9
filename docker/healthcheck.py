"""Container healthcheck: poll /health with retries (parity:
reference docker/healthcheck.py:15-55)."""

import os
import sys
import time
import urllib.request

PORT = os.environ.get("GATEWAY_PORT", "9100")
URL = f"http://127.0.0.1:{PORT}/health"
RETRIES = 3


def main() -> int:
    for attempt in range(RETRIES):
        try:
            with urllib.request.urlopen(URL, timeout=5) as resp:
                if resp.status == 200 and b"ok" in resp.read():
                    return 0
        except Exception as e:
            print(f"healthcheck attempt {attempt + 1}/{RETRIES} failed: {e}", file=sys.stderr)
        time.sleep(2)
    return 1


if __name__ == "__main__":
    sys.exit(main())
