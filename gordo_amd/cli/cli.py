"""CLI — container/operator entrypoints.

Parity: gordo_components/cli/cli.py — click group `gordo-amd` with commands
build, run-server, run-watchman, client (predict/metadata/download-model),
workflow (generate/unique-tags); plus the MI355X-native `fleet` group
(build/serve) replacing the Argo/K8s deployment layer. Config may come from
args or env vars (MACHINE_NAME, MODEL_CONFIG, DATA_CONFIG, OUTPUT_DIR,
MODEL_REGISTER_DIR) exactly like the reference's builder pod contract.
"""
from __future__ import annotations

import json
import logging
import os
import sys

import click
import yaml

from .. import __version__

logger = logging.getLogger(__name__)


@click.group("gordo-amd")
@click.version_option(version=__version__, prog_name="gordo-amd")
@click.option("--log-level", default="INFO")
def gordo(log_level):
    """gordo-amd: build, serve and monitor time-series anomaly models on
    MI355X."""
    logging.basicConfig(
        level=getattr(logging, log_level.upper(), logging.INFO),
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )


# ---------------------------------------------------------------------------
@gordo.command()
@click.option("--name", envvar="MACHINE_NAME", required=True)
@click.option("--model-config", envvar="MODEL_CONFIG", default=None,
              help="YAML/JSON model definition")
@click.option("--data-config", envvar="DATA_CONFIG", required=True,
              help="YAML/JSON dataset config")
@click.option("--metadata", envvar="METADATA", default="{}")
@click.option("--evaluation-config", envvar="EVALUATION_CONFIG", default="{}")
@click.option("--output-dir", envvar="OUTPUT_DIR", required=True)
@click.option("--model-register-dir", envvar="MODEL_REGISTER_DIR", default=None)
@click.option("--print-cv-scores", is_flag=True)
def build(name, model_config, data_config, metadata, evaluation_config,
          output_dir, model_register_dir, print_cv_scores):
    """Train one Machine and dump the artifact dir (builder entrypoint)."""
    from ..builder import ModelBuilder
    from ..machine import Machine

    machine = Machine(
        name=name,
        dataset=yaml.safe_load(data_config),
        model=yaml.safe_load(model_config) if model_config else None,
        metadata=yaml.safe_load(metadata) or {},
        evaluation=yaml.safe_load(evaluation_config) or {},
    )
    model, meta = ModelBuilder(machine).build(
        output_dir=output_dir, model_register_dir=model_register_dir
    )
    if print_cv_scores:
        scores = meta.get("model", {}).get("cross-validation", {}).get("scores", {})
        for k, v in scores.items():
            click.echo(f"{k}: {v.get('mean')}")
    click.echo(f"Built {name} -> {output_dir}")


# ---------------------------------------------------------------------------
@gordo.command("run-server")
@click.option("--model-dir", envvar="MODEL_DIR", required=True)
@click.option("--host", default="0.0.0.0")
@click.option("--port", default=5555, type=int)
@click.option("--workers", default=1, type=int)
def run_server_cmd(model_dir, host, port, workers):
    """Serve one model dir (/prediction, /anomaly/prediction, /metadata...)."""
    from ..server import run_server

    run_server(model_dir, host=host, port=port, workers=workers)


@gordo.command("run-watchman")
@click.option("--project-name", envvar="PROJECT_NAME", required=True)
@click.option("--targets", envvar="TARGETS", required=True,
              help='JSON/YAML map {machine-name: endpoint-url}')
@click.option("--host", default="0.0.0.0")
@click.option("--port", default=5556, type=int)
@click.option("--refresh-interval", default=10.0, type=float)
def run_watchman_cmd(project_name, targets, host, port, refresh_interval):
    """Run the project supervisor (endpoint discovery + health aggregation)."""
    from ..watchman import run_watchman

    run_watchman(project_name, yaml.safe_load(targets), host=host, port=port,
                 refresh_interval=refresh_interval)


# ---------------------------------------------------------------------------
@gordo.group()
def client():
    """Query model servers over a time range."""


@client.command("predict")
@click.argument("start")
@click.argument("end")
@click.option("--watchman-url", default=None)
@click.option("--endpoint", "endpoints", multiple=True)
@click.option("--target", default=None)
@click.option("--output-dir", default=None, help="write parquet per machine")
@click.option("--parallelism", default=10, type=int)
@click.option("--data-provider", default=None,
              help='YAML/JSON provider config, e.g. \'{"type": "RandomDataProvider"}\' '
                   "(parity: the reference CLI's --data-provider)")
def client_predict(start, end, watchman_url, endpoints, target, output_dir,
                   parallelism, data_provider):
    from ..client import Client, ForwardPredictionsToDisk
    from ..dataset.data_provider import GordoBaseDataProvider

    forwarder = ForwardPredictionsToDisk(output_dir) if output_dir else None
    provider = (
        GordoBaseDataProvider.from_dict(yaml.safe_load(data_provider))
        if data_provider
        else None
    )
    c = Client(
        watchman_url=watchman_url,
        endpoints=list(endpoints) or None,
        target=target,
        parallelism=parallelism,
        data_provider=provider,
        prediction_forwarder=forwarder,
    )
    results = c.predict(start, end)
    for r in results:
        n = len(r.predictions) if r.predictions is not None else 0
        click.echo(f"{r.name}: {n} rows, {len(r.error_messages)} errors")


@client.command("metadata")
@click.option("--watchman-url", default=None)
@click.option("--endpoint", "endpoints", multiple=True)
@click.option("--target", default=None)
def client_metadata(watchman_url, endpoints, target):
    from ..client import Client

    c = Client(watchman_url=watchman_url, endpoints=list(endpoints) or None, target=target)
    click.echo(json.dumps(c.get_metadata(), indent=2, default=str))


@client.command("download-model")
@click.option("--watchman-url", default=None)
@click.option("--endpoint", "endpoints", multiple=True)
@click.option("--target", default=None)
@click.option("--output-dir", required=True)
def client_download_model(watchman_url, endpoints, target, output_dir):
    from ..client import Client
    from ..serializer import dump

    c = Client(watchman_url=watchman_url, endpoints=list(endpoints) or None, target=target)
    models = c.download_model()
    os.makedirs(output_dir, exist_ok=True)
    for name, model in models.items():
        dump(model, os.path.join(output_dir, name))
        click.echo(f"{name} -> {output_dir}/{name}")


# ---------------------------------------------------------------------------
@gordo.group()
def workflow():
    """Compile project YAML into a node deployment plan."""


@workflow.command("generate")
@click.option("--machine-config", required=True, type=click.Path(exists=True))
@click.option("--project-name", default=None,
              help="Overrides the YAML's top-level project_name (default: 'project')")
@click.option("--output-dir", default="./models")
@click.option("--model-register-dir", default=None)
@click.option("--n-gpus", default=8, type=int)
@click.option("--output-file", default=None)
def workflow_generate(machine_config, project_name, output_dir, model_register_dir,
                      n_gpus, output_file):
    from ..machine import NormalizedConfig
    from ..workflow import generate_yaml

    with open(machine_config) as f:
        config = NormalizedConfig(yaml.safe_load(f), project_name=project_name)
    text = generate_yaml(
        config,
        project_name=config.project_name,
        output_dir=output_dir,
        registry_dir=model_register_dir,
        n_gpus=n_gpus,
    )
    if output_file:
        with open(output_file, "w") as f:
            f.write(text)
    else:
        click.echo(text)


@workflow.command("unique-tags")
@click.option("--machine-config", required=True, type=click.Path(exists=True))
def workflow_unique_tags(machine_config):
    from ..machine import NormalizedConfig
    from ..workflow import unique_tags

    with open(machine_config) as f:
        config = NormalizedConfig(yaml.safe_load(f))
    for t in unique_tags(config):
        click.echo(t)


# ---------------------------------------------------------------------------
@gordo.group()
def fleet():
    """In-node fleet operations (replaces the reference's Argo/K8s layer)."""


@fleet.command("build")
@click.option("--machine-config", required=True, type=click.Path(exists=True))
@click.option("--project-name", default=None,
              help="Overrides the YAML's top-level project_name (default: 'project')")
@click.option("--output-dir", required=True)
@click.option("--model-register-dir", default=None)
@click.option("--workers", default=None, type=int)
@click.option("--report-file", default=None)
def fleet_build(machine_config, project_name, output_dir, model_register_dir,
                workers, report_file):
    from ..machine import NormalizedConfig
    from ..workflow import FleetBuilder

    with open(machine_config) as f:
        config = NormalizedConfig(yaml.safe_load(f), project_name=project_name)
    builder = FleetBuilder(config, output_dir, model_register_dir, n_workers=workers)
    results = builder.build_all()
    ok = sum(1 for r in results if r["status"] == "succeeded")
    if report_file:
        with open(report_file, "w") as f:
            json.dump(results, f, indent=1, default=str)
    click.echo(f"{ok}/{len(results)} machines built")
    if ok < len(results):
        sys.exit(1)


@fleet.command("serve")
@click.option("--models-dir", required=True, type=click.Path(exists=True))
@click.option("--project-name", default=None,
              help="Defaults to the project name recorded by `fleet build` "
                   "in <models-dir>/project.json, else 'project'")
@click.option("--host", default="0.0.0.0")
@click.option("--port", default=5557, type=int)
def fleet_serve(models_dir, project_name, host, port):
    from ..server.fleet_app import run_fleet_server

    run_fleet_server(models_dir, project_name, host=host, port=port)


if __name__ == "__main__":
    gordo()
