"""
``gordo client`` — CLI over :class:`gordo_amd.client.Client`.

Behavioral spec: the gordo-client package CLI exercised by the
reference's tests/gordo/client/test_client.py:182-394 — subcommands
``predict`` (anomaly predictions over a date range, optionally saved
per-target), ``metadata`` (print or save JSON) and ``download-model``
(serializer layout per target). The workflow template invokes it in
per-machine client pods.
"""
import json
import os
from typing import List

import click

from .custom_types import IsoFormatDateTime
from .. import serializer
from ..client import Client


def make_client(ctx_obj: dict) -> Client:
    """Build the Client from group-level options (tests patch this to
    inject an in-process transport)."""
    return Client(**ctx_obj)


@click.group("client")
@click.option("--project", required=True, help="The project to target")
@click.option("--host", default="localhost", help="The host the server is on")
@click.option("--port", default=443, help="Port the server is on")
@click.option("--scheme", default="https", help="http or https")
@click.option("--parallelism", default=10, help="Concurrent requests")
@click.option(
    "--metadata",
    type=str,
    multiple=True,
    help="key=value pairs to attach to predictions",
)
@click.pass_context
def client_cli(ctx, project, host, port, scheme, parallelism, metadata):
    """Interact with the Gordo ML server's API."""
    meta = dict(kv.split("=", 1) for kv in metadata) if metadata else {}
    ctx.obj = dict(
        project=project,
        host=host,
        port=port,
        scheme=scheme,
        parallelism=parallelism,
        metadata=meta,
    )


@client_cli.command("predict")
@click.argument("start", type=IsoFormatDateTime())
@click.argument("end", type=IsoFormatDateTime())
@click.option(
    "--target", type=str, multiple=True, help="A specific model to predict with"
)
@click.option(
    "--output-dir",
    type=click.Path(exists=True),
    default=None,
    help="Save prediction CSVs (gzip) here, one per target",
)
@click.option(
    "--forward-to-disk",
    type=click.Path(),
    default=None,
    help="Forward each target's predictions as parquet into this dir "
         "(the ForwardPredictionsToDisk forwarder)",
)
@click.pass_context
def predict_cmd(ctx, start, end, target, output_dir, forward_to_disk):
    """Anomaly predictions over [START, END) for each target."""
    from ..client.forwarders import ForwardPredictionsToDisk

    client = make_client(ctx.obj)
    forwarder = (
        ForwardPredictionsToDisk(forward_to_disk) if forward_to_disk else None
    )
    results = client.predict(
        start, end, targets=list(target) or None
    )
    failed: List[str] = []
    for name, frame, errors in results:
        if errors:
            failed.append(name)
            for err in errors:
                click.echo(f"{name}: {err}", err=True)
            continue
        if forwarder is not None:
            forwarder.forward_predictions(frame, name)
        if output_dir:
            path = os.path.join(output_dir, f"{name}.csv.gz")
            frame.to_csv(path, compression="gzip")
            click.echo(f"Saved predictions for {name} -> {path}")
        else:
            click.echo(f"{name}: {len(frame)} predictions")
    if failed:
        raise click.ClickException(
            f"Failed predictions for: {', '.join(failed)}"
        )


@client_cli.command("metadata")
@click.option(
    "--target", type=str, multiple=True, help="Only these models' metadata"
)
@click.option(
    "--output-file",
    type=str,
    default=None,
    help="Save JSON {model: metadata} here instead of printing",
)
@click.pass_context
def metadata_cmd(ctx, target, output_file):
    """Fetch model metadata from the server."""
    client = make_client(ctx.obj)
    metadata = client.get_metadata(targets=list(target) or None)
    payload = json.dumps(metadata, default=str, indent=2)
    if output_file:
        with open(output_file, "w") as f:
            f.write(payload)
        click.echo(f"Saved metadata -> {output_file}")
    else:
        click.echo(payload)


@client_cli.command("download-model")
@click.argument("output-dir", type=click.Path(exists=True))
@click.option(
    "--target", type=str, multiple=True, help="Only download these models"
)
@click.pass_context
def download_model_cmd(ctx, output_dir, target):
    """Download models into OUTPUT_DIR/<name>/ in the serializer layout."""
    client = make_client(ctx.obj)
    models = client.download_model(targets=list(target) or None)
    for name, model in models.items():
        model_dir = os.path.join(output_dir, name)
        os.makedirs(model_dir, exist_ok=True)
        serializer.dump(model, model_dir)
        click.echo(f"Saved model {name} -> {model_dir}")
