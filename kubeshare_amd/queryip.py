"""query-ip init container: writes the node-local scheduler IP to
/kubeshare/library/schedulerIP.txt for the in-container hook to find
(reference cmd/kubeshare-query-ip/main.go:23-35)."""
import os
import sys

from .utils import constants as C


def main(path: str = C.SCHEDULER_IP_FILE) -> int:
    ip = os.environ.get("KUBESHARE_SCHEDULER_IP", "127.0.0.1")
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "w") as f:
        f.write(ip + "\n")
    print(f"schedulerIP.txt <- {ip}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
