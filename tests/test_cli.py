"""CLI tests (parity: tests/.../test_cli.py — CliRunner invoking build /
workflow generate with env vars against tmp dirs)."""
import json
import os

import yaml
from click.testing import CliRunner

from gordo_amd.cli import gordo

DATA_CONFIG = json.dumps(
    {
        "type": "RandomDataset",
        "train_start_date": "2020-01-01T00:00:00Z",
        "train_end_date": "2020-02-01T00:00:00Z",
        "tag_list": ["t1", "t2", "t3", "t4"],
    }
)

MODEL_CONFIG = json.dumps(
    {
        "gordo_amd.model.models.TorchAutoEncoder": {
            "kind": "feedforward_hourglass",
            "epochs": 1,
            "batch_size": 32,
        }
    }
)

PROJECT_YAML = """
globals:
  dataset:
    type: RandomDataset
    train_start_date: "2020-01-01T00:00:00Z"
    train_end_date: "2020-02-01T00:00:00Z"
machines:
  - name: m-a
    dataset:
      tag_list: [x1, x2]
"""


def test_build_command_args(tmp_path):
    runner = CliRunner()
    out = tmp_path / "model"
    result = runner.invoke(
        gordo,
        [
            "build",
            "--name", "cli-machine",
            "--data-config", DATA_CONFIG,
            "--model-config", MODEL_CONFIG,
            "--output-dir", str(out),
            "--print-cv-scores",
        ],
    )
    assert result.exit_code == 0, result.output
    assert (out / "model.pkl").exists()
    assert (out / "metadata.json").exists()
    assert "explained_variance_score" in result.output


def test_build_command_env_vars(tmp_path):
    runner = CliRunner()
    out = tmp_path / "model-env"
    env = {
        "MACHINE_NAME": "env-machine",
        "DATA_CONFIG": DATA_CONFIG,
        "MODEL_CONFIG": MODEL_CONFIG,
        "OUTPUT_DIR": str(out),
        "MODEL_REGISTER_DIR": str(tmp_path / "registry"),
    }
    result = runner.invoke(gordo, ["build"], env=env)
    assert result.exit_code == 0, result.output
    assert (out / "model.pkl").exists()
    # cache registry written
    assert os.listdir(tmp_path / "registry")


def test_workflow_generate(tmp_path):
    cfg = tmp_path / "project.yaml"
    cfg.write_text(PROJECT_YAML)
    runner = CliRunner()
    result = runner.invoke(
        gordo,
        ["workflow", "generate", "--machine-config", str(cfg), "--project-name", "p1"],
    )
    assert result.exit_code == 0, result.output
    plan = yaml.safe_load(result.output)
    assert plan["kind"] == "NodePlan"
    assert plan["metadata"]["project-name"] == "p1"


def test_workflow_unique_tags(tmp_path):
    cfg = tmp_path / "project.yaml"
    cfg.write_text(PROJECT_YAML)
    runner = CliRunner()
    result = runner.invoke(gordo, ["workflow", "unique-tags", "--machine-config", str(cfg)])
    assert result.exit_code == 0
    assert result.output.split() == ["x1", "x2"]


def test_fleet_build_command(tmp_path):
    cfg = tmp_path / "project.yaml"
    cfg.write_text(PROJECT_YAML)
    runner = CliRunner()
    report = tmp_path / "report.json"
    result = runner.invoke(
        gordo,
        [
            "fleet", "build",
            "--machine-config", str(cfg),
            "--output-dir", str(tmp_path / "models"),
            "--workers", "1",
            "--report-file", str(report),
        ],
    )
    assert result.exit_code == 0, result.output
    rep = json.loads(report.read_text())
    assert rep[0]["status"] == "succeeded"


def test_version_and_help_surfaces():
    """Every reference CLI surface parses: --version and each command group's
    --help exit 0 (run-server/run-watchman/client start servers or need
    endpoints, so only their arg parsing is smoked here)."""
    runner = CliRunner()
    r = runner.invoke(gordo, ["--version"])
    assert r.exit_code == 0 and "gordo-amd" in r.output
    for args in (
        ["build", "--help"],
        ["run-server", "--help"],
        ["run-watchman", "--help"],
        ["client", "--help"],
        ["client", "predict", "--help"],
        ["client", "metadata", "--help"],
        ["client", "download-model", "--help"],
        ["workflow", "generate", "--help"],
        ["fleet", "build", "--help"],
        ["fleet", "serve", "--help"],
    ):
        r = runner.invoke(gordo, args)
        assert r.exit_code == 0, (args, r.output)


def test_client_predict_data_provider_option(monkeypatch):
    """--data-provider YAML/JSON reaches Client as a constructed provider
    (reference CLI parity: predictions can be pulled without library code)."""
    import gordo_amd.client as client_mod
    from gordo_amd.dataset.data_provider import RandomDataProvider

    captured = {}

    class FakeClient:
        def __init__(self, **kwargs):
            captured.update(kwargs)

        def predict(self, start, end):
            return []

    monkeypatch.setattr(client_mod, "Client", FakeClient)
    runner = CliRunner()
    result = runner.invoke(
        gordo,
        [
            "client", "predict",
            "--endpoint", "http://127.0.0.1:1",
            "--data-provider", '{"type": "RandomDataProvider", "min_size": 5, "max_size": 7}',
            "2020-01-01T00:00:00Z", "2020-01-02T00:00:00Z",
        ],
    )
    assert result.exit_code == 0, result.output
    provider = captured["data_provider"]
    assert isinstance(provider, RandomDataProvider)
    assert (provider.min_size, provider.max_size) == (5, 7)


def test_build_command_bad_config_fails_with_message(tmp_path):
    """Builder-pod error reporting (reference: the build entrypoint surfaces
    config errors with a non-zero exit instead of a silent crash)."""
    runner = CliRunner()
    result = runner.invoke(
        gordo,
        [
            "build",
            "--name", "broken",
            "--data-config", '{"type": "RandomDataset", "tag_list": ["t0"], '
            '"train_start_date": "2020-01-01T00:00:00Z", '
            '"train_end_date": "2020-02-01T00:00:00Z"}',
            "--model-config", '{"no.such.module.Model": {}}',
            "--output-dir", str(tmp_path / "out"),
        ],
    )
    assert result.exit_code != 0
    # the failure reason reaches the operator (import error of the dotted path)
    blob = (result.output or "") + str(result.exception)
    assert "no.such" in blob or "No module" in blob
