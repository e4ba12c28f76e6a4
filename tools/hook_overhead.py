"""Dispatch hot-path overhead: ns/hipLaunchKernel bare vs under the
LD_PRELOAD gate with a valid lease (CPU, fake-HIP substrate; numbers
in profiles/hook_hotpath_overhead.txt)."""
import os
import socket
import subprocess
import sys
import tempfile
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
TL = os.path.join(REPO, "native", "testlibs")


def main():
    subprocess.run(["make", "-C", os.path.join(REPO, "native"),
                    "testlibs"], check=True, capture_output=True)
    rate = os.path.join(TL, "launch_rate")
    env0 = dict(os.environ, LD_LIBRARY_PATH=TL)
    print("bare :", subprocess.run([rate], env=env0, capture_output=True,
                                   text=True).stdout.strip())
    tmp = tempfile.mkdtemp()
    cfg = os.path.join(tmp, "c")
    os.makedirs(cfg)
    open(os.path.join(cfg, "GPU-x"), "w").write("1\nmb/pod 1.0 1.0 0\n")
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    schd = subprocess.Popen(
        [os.path.join(REPO, "native", "gpu-schd"), "-p", cfg, "-f",
         "GPU-x", "-P", str(port), "-q", "10000", "-m", "20",
         "-w", "60000"], stderr=subprocess.DEVNULL)
    time.sleep(0.4)
    try:
        env1 = dict(env0,
                    LD_PRELOAD=os.path.join(REPO, "native",
                                            "libhiphook.so"),
                    SCHEDULER_IP="127.0.0.1", SCHEDULER_PORT=str(port),
                    POD_NAME="mb/pod")
        print("gated:", subprocess.run([rate], env=env1,
                                       capture_output=True,
                                       text=True).stdout.strip())
    finally:
        schd.terminate()
        schd.wait()
    return 0


if __name__ == "__main__":
    sys.exit(main())
