# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Multi-process serving workers + L4 round-robin proxy.

Parity target: nuclio worker replicas (reference ServingRuntime
`min_replicas/max_replicas`, SURVEY §2.4 strategy 5).  One python
process caps the HTTP layer at ~100 req/s regardless of engine speed
(measured, profiles/README.md) — worker PROCESSES remove the GIL from
the serving path.  Each worker rebuilds the graph from the function
spec and serves on its own port; the parent runs a byte-level
round-robin TCP proxy (connections pin to a backend, so keep-alive
clients stay pinned; 288 GB HBM fits several full model replicas).

Child entry: ``python -m mlrun_amd.serving.workers <spec.yaml> <port>``
"""


import os
import socket
import subprocess
import sys
import time
import typing

from ..utils import logger


def _free_port() -> int:
    with socket.socket() as sock:
        sock.bind(("127.0.0.1", 0))
        return sock.getsockname()[1]


class TcpRoundRobinProxy:
    """Minimal L4 proxy: each accepted connection is piped to the next
    backend (asyncio streams in a daemon thread)."""

    def __init__(self, backends: typing.List[typing.Tuple[str, int]],
                 host="127.0.0.1", port: int = 0):
        self.backends = backends
        self.host = host
        self.port = port or _free_port()
        self._thread = None
        self._loop = None
        self._next = 0
        self.active = 0  # live proxied connections (autoscale signal)

    def add_backend(self, host: str, port: int):
        """Register a new worker (autoscale-up); next connections
        round-robin over the larger set."""
        self.backends.append((host, port))

    @property
    def address(self) -> str:
        return f"http://{self.host}:{self.port}"

    def start(self):
        import asyncio
        import threading

        ready = threading.Event()

        def run():
            loop = asyncio.new_event_loop()
            asyncio.set_event_loop(loop)
            self._loop = loop

            async def pipe(reader, writer):
                try:
                    while True:
                        data = await reader.read(65536)
                        if not data:
                            break
                        writer.write(data)
                        await writer.drain()
                except (ConnectionError, asyncio.CancelledError):
                    pass
                finally:
                    try:
                        writer.close()
                    except Exception:
                        pass

            async def handle(client_r, client_w):
                host, port = self.backends[self._next %
                                           len(self.backends)]
                self._next += 1
                try:
                    backend_r, backend_w = await asyncio.open_connection(
                        host, port)
                except OSError:
                    client_w.close()
                    return
                self.active += 1
                try:
                    await asyncio.gather(pipe(client_r, backend_w),
                                         pipe(backend_r, client_w))
                finally:
                    self.active -= 1

            async def main():
                server = await asyncio.start_server(handle, self.host,
                                                    self.port)
                ready.set()
                async with server:
                    await server.serve_forever()

            try:
                loop.run_until_complete(main())
            except Exception:
                pass
            finally:
                # drain pending pipe tasks quietly on stop
                pending = [t for t in asyncio.all_tasks(loop)
                           if not t.done()]
                for task in pending:
                    task.cancel()
                if pending:
                    loop.run_until_complete(asyncio.gather(
                        *pending, return_exceptions=True))
                loop.close()

        self._thread = __import__("threading").Thread(target=run,
                                                      daemon=True)
        self._thread.start()
        ready.wait(timeout=10)

    def stop(self):
        if self._loop is not None:
            self._loop.call_soon_threadsafe(self._loop.stop)


class WorkerPool:
    """N serving-host processes + proxy, built from a function spec."""

    def __init__(self, function_spec: dict, workers: int,
                 env: dict = None, max_workers: int = 0,
                 scale_connections_per_worker: int = None,
                 scale_interval: float = 2.0):
        from ..config import config

        if scale_connections_per_worker is None:
            scale_connections_per_worker = int(
                config.runtimes.serving.autoscale_connections_per_worker)
        self.spec = function_spec
        self.workers = workers
        self.max_workers = max(max_workers, workers)
        # scale-up trigger: live proxied connections per worker above
        # this -> add a worker (nuclio min/max_replicas analog)
        self.scale_connections_per_worker = scale_connections_per_worker
        self.scale_interval = scale_interval
        self.env = env or {}
        self.processes: typing.List[subprocess.Popen] = []
        self.ports: typing.List[int] = []
        self.proxy: typing.Optional[TcpRoundRobinProxy] = None
        self._spec_path = ""
        self._scaler = None
        self._stopping = False

    @property
    def address(self) -> str:
        return self.proxy.address

    def start(self, timeout: float = 180) -> str:
        import tempfile

        import yaml

        spec_file = tempfile.NamedTemporaryFile(
            "w", suffix=".yaml", delete=False)
        yaml.safe_dump(self.spec, spec_file)
        spec_file.close()
        env = dict(os.environ)
        repo_root = os.path.dirname(os.path.dirname(
            os.path.dirname(os.path.abspath(__file__))))
        env["PYTHONPATH"] = repo_root + os.pathsep + \
            env.get("PYTHONPATH", "")
        env.update({k: str(v) for k, v in self.env.items()})
        self._env = env
        self._spec_path = spec_file.name
        for _ in range(self.workers):
            self._spawn_worker()
        deadline = time.time() + timeout
        import requests

        for port in self.ports:
            while True:
                try:
                    requests.get(f"http://127.0.0.1:{port}/healthz",
                                 timeout=2)
                    break
                except Exception:
                    if time.time() > deadline:
                        self.stop()
                        raise TimeoutError(
                            f"serving worker on port {port} did not "
                            f"become ready")
                    time.sleep(0.5)
        self.proxy = TcpRoundRobinProxy(
            [("127.0.0.1", p) for p in self.ports])
        self.proxy.start()
        if self.max_workers > self.workers:
            import threading

            self._scaler = threading.Thread(target=self._scale_loop,
                                            daemon=True,
                                            name="worker-autoscaler")
            self._scaler.start()
        logger.info("serving worker pool started",
                    workers=self.workers, max_workers=self.max_workers,
                    address=self.address)
        return self.address

    def _spawn_worker(self) -> int:
        port = _free_port()
        self.ports.append(port)
        proc = subprocess.Popen(
            [sys.executable, "-m", "mlrun_amd.serving.workers",
             self._spec_path, str(port)], env=self._env)
        self.processes.append(proc)
        return port

    def _scale_loop(self):
        """Scale-up monitor (nuclio max_replicas analog): when live
        proxied connections exceed the per-worker budget, start
        another worker and add it to the round-robin set.  No
        scale-down — decode engines hold captured graphs and HBM is
        plentiful (288 GB); idle workers cost nothing hot."""
        import requests

        while not self._stopping and \
                len(self.ports) < self.max_workers:
            time.sleep(self.scale_interval)
            if self.proxy is None:
                continue
            threshold = self.scale_connections_per_worker * \
                len(self.ports)
            if self.proxy.active <= threshold:
                continue
            port = self._spawn_worker()
            deadline = time.time() + 180
            while not self._stopping:
                try:
                    requests.get(f"http://127.0.0.1:{port}/healthz",
                                 timeout=2)
                    break
                except Exception:
                    if time.time() > deadline:
                        logger.warning("autoscaled worker never became "
                                       "ready", port=port)
                        return
                    time.sleep(0.5)
            self.proxy.add_backend("127.0.0.1", port)
            logger.info("worker pool scaled up", workers=len(self.ports),
                        active_connections=self.proxy.active)

    def stop(self):
        self._stopping = True
        if self.proxy is not None:
            self.proxy.stop()
        if self._spec_path and os.path.exists(self._spec_path):
            os.unlink(self._spec_path)
        for proc in self.processes:
            if proc.poll() is None:
                proc.terminate()
        for proc in self.processes:
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()
        self.processes = []


def _worker_main(spec_path: str, port: int):
    """Child process: rebuild the serving function and host it."""
    import yaml

    import mlrun_amd
    from .server import GraphServerHost

    with open(spec_path) as stream:
        spec = yaml.safe_load(stream)
    fn = mlrun_amd.new_function(runtime=spec)
    server = fn._build_server(None)
    host = GraphServerHost(server, port=port)
    host.start()
    logger.info("serving worker ready", port=port, pid=os.getpid())
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    _worker_main(sys.argv[1], int(sys.argv[2]))
