# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Python client + lifecycle manager for the C++ log-collector daemon
(mlrun_amd/native/log_collector.cpp — the reference's Go gRPC service
rebuilt in C++ over a newline-JSON TCP protocol)."""

import json
import os
import socket
import subprocess
import time
import typing

from ..config import config
from ..errors import MLRunRuntimeError

NATIVE_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "native")
BINARY = os.path.join(NATIVE_DIR, "log_collector")
SOURCE = os.path.join(NATIVE_DIR, "log_collector.cpp")


def build_binary(force: bool = False) -> str:
    """Compile the daemon with g++ (done by __graft_entry__.build)."""
    if os.path.isfile(BINARY) and not force and \
            os.path.getmtime(BINARY) >= os.path.getmtime(SOURCE):
        return BINARY
    cmd = ["g++", "-O2", "-std=c++17", "-pthread", SOURCE, "-o", BINARY]
    result = subprocess.run(cmd, capture_output=True, text=True)
    if result.returncode != 0:
        raise MLRunRuntimeError(
            f"log_collector build failed: {result.stderr}")
    return BINARY


class LogCollectorClient:
    """Speaks the daemon protocol; can also own the daemon process."""

    def __init__(self, port: int = 0, log_dir: str = None,
                 start_daemon: bool = True):
        self.port = port or _free_port()
        self.log_dir = log_dir or os.path.join(config.base_dir,
                                               "collected-logs")
        self._process: typing.Optional[subprocess.Popen] = None
        if start_daemon:
            self.start_daemon()

    def start_daemon(self, timeout: float = 10.0):
        build_binary()
        os.makedirs(self.log_dir, exist_ok=True)
        self._process = subprocess.Popen(
            [BINARY, str(self.port), self.log_dir],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            try:
                self._call({"op": "list_runs_in_progress"})
                return self
            except (ConnectionError, OSError):
                time.sleep(0.05)
        raise MLRunRuntimeError("log_collector daemon did not start")

    def _connect(self) -> socket.socket:
        sock = socket.create_connection(("127.0.0.1", self.port), timeout=10)
        return sock

    def _call(self, request: dict) -> dict:
        with self._connect() as sock:
            sock.sendall((json.dumps(request) + "\n").encode())
            resp = self._read_line(sock)
            return json.loads(resp)

    @staticmethod
    def _read_line(sock: socket.socket) -> bytes:
        chunks = []
        while True:
            byte = sock.recv(1)
            if not byte or byte == b"\n":
                break
            chunks.append(byte)
        return b"".join(chunks)

    # --- the 6 operations (parity: proto/log_collector.proto:21-28) ---
    def start_log(self, run_uid: str, project: str, source: str) -> bool:
        return self._call({"op": "start_log", "run_uid": run_uid,
                           "project": project, "source": source})["success"]

    def get_log_size(self, run_uid: str, project: str) -> int:
        return int(self._call({"op": "get_log_size", "run_uid": run_uid,
                               "project": project})["size"])

    def get_logs(self, run_uid: str, project: str, offset: int = 0,
                 size: int = 0) -> bytes:
        with self._connect() as sock:
            sock.sendall((json.dumps(
                {"op": "get_logs", "run_uid": run_uid, "project": project,
                 "offset": offset, "size": size}) + "\n").encode())
            header = json.loads(self._read_line(sock))
            if not header.get("success"):
                raise MLRunRuntimeError(header.get("error", "get_logs"))
            want = int(header["size"])
            data = b""
            while len(data) < want:
                chunk = sock.recv(min(65536, want - len(data)))
                if not chunk:
                    break
                data += chunk
            return data

    def stop_logs(self, project: str, run_uid: str = "") -> bool:
        return self._call({"op": "stop_logs", "run_uid": run_uid,
                           "project": project})["success"]

    def delete_logs(self, run_uid: str, project: str) -> bool:
        return self._call({"op": "delete_logs", "run_uid": run_uid,
                           "project": project})["success"]

    def list_runs_in_progress(self) -> list:
        return self._call({"op": "list_runs_in_progress"})["runs"]

    def shutdown(self):
        try:
            self._call({"op": "shutdown"})
        except Exception:
            pass
        if self._process is not None:
            try:
                self._process.wait(timeout=5)
            except subprocess.TimeoutExpired:
                self._process.kill()
            self._process = None

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.shutdown()


def _free_port() -> int:
    sock = socket.socket()
    sock.bind(("127.0.0.1", 0))
    port = sock.getsockname()[1]
    sock.close()
    return port
