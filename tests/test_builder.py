# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Function image builder (node-local kaniko replacement)."""

import os
import subprocess
import sys

import yaml

import mlrun_amd
from mlrun_amd.utils.builder import build_image, build_runtime


class TestBuildImage:
    def test_materializes_source_requirements_and_manifest(self, tmp_path):
        dest = str(tmp_path / "img")
        build_image(dest,
                    source_files={"fn.py": "print('hi from image')\n",
                                  "pkg/util.py": "X = 1\n"},
                    requirements=["numpy", "pyyaml"],
                    commands=["export MY_FLAG=1"],
                    command="fn.py")
        assert os.path.exists(os.path.join(dest, "pkg/util.py"))
        manifest = yaml.safe_load(open(os.path.join(dest, "image.yaml")))
        assert manifest["state"] == "ready"
        assert manifest["requirements"] == ["numpy", "pyyaml"]
        reqs = open(os.path.join(dest, "requirements.txt")).read()
        assert "numpy" in reqs

    def test_run_sh_executes_the_entrypoint(self, tmp_path):
        dest = str(tmp_path / "img")
        build_image(dest, source_files={"fn.py": "print(6 * 7)\n"},
                    command="fn.py")
        out = subprocess.run(["sh", os.path.join(dest, "run.sh")],
                             capture_output=True, text=True)
        assert out.returncode == 0 and out.stdout.strip() == "42"

    def test_copies_existing_source_path(self, tmp_path):
        src = tmp_path / "code.py"
        src.write_text("VALUE = 9\n")
        dest = str(tmp_path / "img")
        build_image(dest, source_files={"code.py": str(src)})
        assert open(os.path.join(dest, "code.py")).read() == "VALUE = 9\n"


class TestBuildRuntime:
    def test_job_deploy_builds_image_and_runs(self):
        fn = mlrun_amd.new_function(name="builder-fn", kind="job")
        fn.with_code(body="def handler(context):\n"
                          "    context.log_result('out', 11)\n")
        assert fn.deploy()
        assert fn.spec.image and os.path.isdir(fn.spec.image)
        assert fn.status.state == "ready"
        run = fn.run(handler="handler", local=True)
        assert run.outputs["out"] == 11

    def test_build_runtime_walks_source_dir(self, tmp_path):
        (tmp_path / "mod.py").write_text("Y = 3\n")
        fn = mlrun_amd.new_function(name="srcdir-fn", kind="job")
        fn.spec.build["source"] = str(tmp_path)
        build_runtime(fn)
        assert os.path.exists(os.path.join(fn.spec.image, "mod.py"))

    def test_cli_build_command(self, tmp_path):
        code = tmp_path / "cli_fn.py"
        code.write_text("def handler(context):\n    return 1\n")
        from mlrun_amd.__main__ import main

        main(["build", "--name", "cli-built", "--command", str(code)])
        from mlrun_amd.config import config

        image = os.path.join(config.base_dir, "images", "default",
                             "cli-built", "latest")
        assert os.path.exists(os.path.join(image, "image.yaml"))
