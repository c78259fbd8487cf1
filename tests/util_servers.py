"""Test fixtures: spawn the C++ modelxd / modelx-s3d binaries on loopback."""
import os
import shutil
import socket
import subprocess
import time

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
MODELXD = os.path.join(REPO_ROOT, "bin", "modelxd")
S3D = os.path.join(REPO_ROOT, "bin", "modelx-s3d")

ACCESS_KEY = "modelx"
SECRET_KEY = "modelx123"
BUCKET = "modelx"


def _build_servers():
    if not (os.path.exists(MODELXD) and os.path.exists(S3D)):
        subprocess.run(["make", "servers"], cwd=REPO_ROOT, check=True, capture_output=True)


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def wait_http(port: int, path: str = "/healthz", timeout: float = 10.0):
    import requests

    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            r = requests.get(f"http://127.0.0.1:{port}{path}", timeout=1)
            if r.status_code < 500:
                return
        except requests.RequestException:
            pass
        time.sleep(0.05)
    raise TimeoutError(f"server on port {port} did not come up")


class ServerProc:
    def __init__(self, args, port):
        self.proc = subprocess.Popen(args, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        self.port = port
        self.url = f"http://127.0.0.1:{port}"

    def stop(self):
        self.proc.terminate()
        try:
            self.proc.wait(timeout=5)
        except subprocess.TimeoutExpired:
            self.proc.kill()


def start_modelxd_local(data_dir: str) -> ServerProc:
    _build_servers()
    port = free_port()
    p = ServerProc([MODELXD, "--listen", f"127.0.0.1:{port}", "--local-data", data_dir], port)
    wait_http(port)
    return p


def start_s3d(root_dir: str, port: int = 0) -> ServerProc:
    _build_servers()
    port = port or free_port()
    p = ServerProc(
        [S3D, "--listen", f"127.0.0.1:{port}", "--root", root_dir,
         "--access-key", ACCESS_KEY, "--secret-key", SECRET_KEY],
        port,
    )
    wait_http(port)
    os.makedirs(os.path.join(root_dir, BUCKET), exist_ok=True)
    return p


def start_modelxd_s3(s3_url: str, redirect: bool = True, port: int = 0) -> ServerProc:
    _build_servers()
    port = port or free_port()
    args = [MODELXD, "--listen", f"127.0.0.1:{port}", "--s3-url", s3_url,
            "--s3-bucket", BUCKET, "--s3-access-key", ACCESS_KEY,
            "--s3-secret-key", SECRET_KEY]
    if redirect:
        args.append("--enable-redirect")
    p = ServerProc(args, port)
    wait_http(port)
    return p
