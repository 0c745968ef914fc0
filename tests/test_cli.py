"""`pio` CLI tests (reference: tools Console + commands behaviors,
integration scenarios in tests/pio_tests/scenarios/basic_app_usecases.py)."""

import json

import pytest
from click.testing import CliRunner

from predictionio_amd.cli.main import cli


@pytest.fixture()
def runner(mem_storage):
    return CliRunner()


class TestAppCommands:
    def test_app_lifecycle(self, runner):
        r = runner.invoke(cli, ["app", "new", "cliapp",
                                "--access-key", "KEY1"])
        assert r.exit_code == 0 and "Access Key: KEY1" in r.output
        r = runner.invoke(cli, ["app", "new", "cliapp"])
        assert r.exit_code == 1 and "already exists" in r.output
        r = runner.invoke(cli, ["app", "list"])
        assert "cliapp" in r.output and "KEY1" in r.output
        r = runner.invoke(cli, ["app", "show", "cliapp"])
        assert "App ID" in r.output
        r = runner.invoke(cli, ["app", "delete", "cliapp", "-f"])
        assert r.exit_code == 0
        r = runner.invoke(cli, ["app", "show", "cliapp"])
        assert r.exit_code == 1

    def test_channels(self, runner):
        runner.invoke(cli, ["app", "new", "chapp"])
        r = runner.invoke(cli, ["app", "channel-new", "chapp", "chan-1"])
        assert r.exit_code == 0
        r = runner.invoke(cli, ["app", "channel-new", "chapp", "bad name!"])
        assert r.exit_code == 1
        r = runner.invoke(cli, ["app", "show", "chapp"])
        assert "chan-1" in r.output
        r = runner.invoke(cli, ["app", "channel-delete", "chapp",
                                "chan-1", "-f"])
        assert r.exit_code == 0

    def test_accesskeys(self, runner):
        runner.invoke(cli, ["app", "new", "akapp", "--access-key", "K0"])
        r = runner.invoke(cli, ["accesskey", "new", "akapp", "buy", "view",
                                "--access-key", "K1"])
        assert r.exit_code == 0
        r = runner.invoke(cli, ["accesskey", "list", "akapp"])
        assert "K1" in r.output and "buy,view" in r.output
        assert runner.invoke(cli, ["accesskey", "delete", "K1"]).exit_code \
            == 0
        assert runner.invoke(cli, ["accesskey", "delete", "K1"]).exit_code \
            == 1


class TestTrainDeployFlow:
    def test_version(self, runner):
        r = runner.invoke(cli, ["version"])
        assert r.exit_code == 0 and r.output.strip()

    def test_train_from_engine_dir(self, runner, tmp_path):
        variant = {"id": "clitest",
                   "engineFactory": "tests.fake_engine.JsonEngineFactory",
                   "datasource": {"params": {"n": 4}},
                   "algorithms": [{"name": "", "params": {}}]}
        (tmp_path / "engine.json").write_text(json.dumps(variant))
        r = runner.invoke(cli, ["train", "--engine-dir", str(tmp_path)])
        assert r.exit_code == 0, r.output
        assert "Training completed" in r.output

    def test_import_export(self, runner, tmp_path):
        runner.invoke(cli, ["app", "new", "ioapp"])
        evs = [{"event": "rate", "entityType": "user",
                "entityId": f"u{i}", "targetEntityType": "item",
                "targetEntityId": "i1", "properties": {"rating": i}}
               for i in range(4)]
        f = tmp_path / "in.json"
        f.write_text("\n".join(json.dumps(e) for e in evs))
        # find the app id from `app show` listing via storage
        from predictionio_amd.data import storage
        app_id = storage.get_meta_data_apps().get_by_name("ioapp").id
        r = runner.invoke(cli, ["import", "--appid", str(app_id),
                                "--input", str(f)])
        assert r.exit_code == 0 and "Imported 4" in r.output
        out = tmp_path / "out.json"
        r = runner.invoke(cli, ["export", "--appid", str(app_id),
                                "--output", str(out)])
        assert r.exit_code == 0 and "Exported 4" in r.output
        # channel round-trip: import into a named channel, export only it
        runner.invoke(cli, ["app", "channel-new", "ioapp", "ch1"])
        r = runner.invoke(cli, ["import", "--appid", str(app_id),
                                "--channel", "ch1", "--input", str(f)])
        assert r.exit_code == 0, r.output
        out2 = tmp_path / "out2.json"
        r = runner.invoke(cli, ["export", "--appid", str(app_id),
                                "--channel", "ch1",
                                "--output", str(out2)])
        assert r.exit_code == 0 and "Exported 4" in r.output
        got = [json.loads(l) for l in out2.read_text().splitlines()]
        assert len(got) == 4
        assert {e["entityId"] for e in got} == {f"u{i}" for i in range(4)}

    def test_template_list(self, runner):
        r = runner.invoke(cli, ["template", "list"])
        assert "recommendation" in r.output

    def test_template_get(self, runner, tmp_path):
        dst = tmp_path / "myengine"
        r = runner.invoke(cli, ["template", "get", "classification",
                                str(dst)])
        assert r.exit_code == 0
        assert (dst / "engine.json").exists()

    def test_eval_cmd(self, runner, mem_storage):
        r = runner.invoke(cli, ["eval",
                                "tests.fake_engine_eval.ZeroEvaluation"])
        assert r.exit_code == 0, r.output
        assert "Best score" in r.output


class TestStatus:
    def test_status_healthy(self, runner):
        r = runner.invoke(cli, ["status"])
        assert r.exit_code == 0, r.output
        assert "ready to go" in r.output


class TestRunCommand:
    def test_run_arbitrary_main(self, runner):
        """`pio run pkg.mod.fn args...` (commands/Engine.run + FakeWorkflow
        parity: run an arbitrary entry point under the PIO environment)."""
        from tests import fake_engine
        fake_engine.RUN_CALLS.clear()
        r = runner.invoke(cli, ["run", "tests.fake_engine.fake_main",
                                "a", "b"])
        assert r.exit_code == 0, r.output
        assert fake_engine.RUN_CALLS == [("a", "b")]


class TestTemplateMinVersion:
    def test_build_rejects_newer_min_version(self, tmp_path):
        import json
        from click.testing import CliRunner
        from predictionio_amd.cli.main import cli
        d = tmp_path / "eng"
        d.mkdir()
        (d / "template.json").write_text(
            json.dumps({"pio": {"version": {"min": "99.0.0"}}}))
        r = CliRunner().invoke(cli, ["build", "--engine-dir", str(d)])
        assert r.exit_code == 1
        assert "requires PIO" in r.output

    def test_train_rejects_newer_min_version(self, tmp_path):
        """The reference checks the gate on train too
        (commands/Engine.scala:188-190)."""
        import json
        from click.testing import CliRunner
        from predictionio_amd.cli.main import cli
        d = tmp_path / "eng"
        d.mkdir()
        (d / "template.json").write_text(
            json.dumps({"pio": {"version": {"min": "99.0.0"}}}))
        r = CliRunner().invoke(cli, ["train", "--engine-dir", str(d)])
        assert r.exit_code == 1
        assert "requires PIO" in r.output
