# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""C++ log-collector daemon tests (the analog of the reference's Go
logcollector_test.go tier, run against the real daemon)."""

import os
import time

import pytest

from mlrun_amd.utils.log_collector import LogCollectorClient, build_binary


@pytest.fixture(scope="module")
def daemon(tmp_path_factory):
    log_dir = str(tmp_path_factory.mktemp("collected"))
    client = LogCollectorClient(log_dir=log_dir)
    yield client
    client.shutdown()


def _wait_for_size(client, uid, project, minimum, timeout=5.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if client.get_log_size(uid, project) >= minimum:
            return True
        time.sleep(0.05)
    return False


class TestLogCollector:
    def test_build(self):
        binary = build_binary()
        assert os.path.isfile(binary)

    def test_start_and_collect(self, daemon, tmp_path):
        source = tmp_path / "live.log"
        source.write_text("first line\n")
        assert daemon.start_log("u1", "p", str(source))
        assert _wait_for_size(daemon, "u1", "p", len("first line\n"))
        # append and see it follow
        with open(source, "a") as fp:
            fp.write("second line\n")
        assert _wait_for_size(daemon, "u1", "p", len("first line\nsecond "
                                                     "line\n"))
        logs = daemon.get_logs("u1", "p")
        assert logs == b"first line\nsecond line\n"

    def test_offset_and_size(self, daemon, tmp_path):
        source = tmp_path / "live2.log"
        source.write_text("0123456789")
        daemon.start_log("u2", "p", str(source))
        assert _wait_for_size(daemon, "u2", "p", 10)
        assert daemon.get_logs("u2", "p", offset=3) == b"3456789"
        assert daemon.get_logs("u2", "p", offset=2, size=4) == b"2345"

    def test_list_stop_delete(self, daemon, tmp_path):
        source = tmp_path / "live3.log"
        source.write_text("x")
        daemon.start_log("u3", "proj3", str(source))
        assert _wait_for_size(daemon, "u3", "proj3", 1)
        assert "proj3/u3" in daemon.list_runs_in_progress()
        daemon.stop_logs("proj3", "u3")
        time.sleep(0.2)
        assert "proj3/u3" not in daemon.list_runs_in_progress()
        # log file still readable after stop
        assert daemon.get_logs("u3", "proj3") == b"x"
        daemon.delete_logs("u3", "proj3")
        with pytest.raises(Exception):
            daemon.get_logs("u3", "proj3")

    def test_missing_log(self, daemon):
        with pytest.raises(Exception):
            daemon.get_logs("nope", "p")
        assert daemon.get_log_size("nope", "p") == -1

    def test_list_runs_in_progress(self, daemon, tmp_path):
        src = tmp_path / "live.log"
        src.write_text("streaming\n")
        daemon.start_log("uid-live", "proj-x", str(src))
        import time

        deadline = time.time() + 5
        while time.time() < deadline:
            runs = daemon.list_runs_in_progress() \
                if hasattr(daemon, "list_runs_in_progress") else \
                daemon._call({"op": "list_runs_in_progress"}).get(
                    "runs", [])
            if any("uid-live" in str(r) for r in runs):
                break
            time.sleep(0.2)
        assert any("uid-live" in str(r) for r in runs)
        daemon.stop_logs("proj-x", "uid-live")
