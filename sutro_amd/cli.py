"""`sutro` CLI: command mirror of the SDK (reference `/root/reference/sutro/cli.py`).

Commands: login, jobs {list,status,results,cancel,attach}, datasets
{create,list,files,upload,download}, cache {clear,show}, quotas, models,
set-base-url, serve, bench-info.
"""

from __future__ import annotations

import json
from typing import Optional

import click

from .validation import load_config, save_config


def _client():
    from .sdk import Sutro

    return Sutro()


@click.group()
def cli():
    """Sutro-AMD: MI355X-native batch inference."""


@cli.command()
@click.option("--api-key", prompt=True, hide_input=True)
def login(api_key: str):
    """Store an API key in ~/.sutro/config.json."""
    config = load_config()
    config["api_key"] = api_key
    save_config(config)
    ok = _client().try_authentication(api_key).get("authenticated", False)
    click.echo("✔ Authenticated" if ok else "✗ Authentication failed")


@cli.command("set-base-url")
@click.argument("base_url")
def set_base_url(base_url: str):
    """Point the client at a service URL (or 'local')."""
    config = load_config()
    config["base_url"] = base_url
    save_config(config)
    click.echo(f"base_url set to {base_url}")


@cli.group()
def jobs():
    """Job operations."""


@jobs.command("list")
def jobs_list():
    df = _client().list_jobs()
    if len(df) == 0:
        click.echo("no jobs")
        return
    cols = [c for c in ("job_id", "status", "num_rows", "input_tokens",
                        "output_tokens", "job_cost", "datetime_created")
            if c in df.columns]
    click.echo(df[cols].to_string(index=False))


@jobs.command("status")
@click.argument("job_id")
def jobs_status(job_id: str):
    click.echo(_client().get_job_status(job_id))


@jobs.command("results")
@click.argument("job_id")
@click.option("--save", type=click.Path(), default=None,
              help="write results to a file")
@click.option("--save-format", type=click.Choice(["parquet", "csv", "json"]),
              default="parquet")
@click.option("--include-inputs", is_flag=True)
def jobs_results(job_id: str, save: Optional[str], save_format: str,
                 include_inputs: bool):
    df = _client().get_job_results(job_id, include_inputs=include_inputs)
    if save:
        if save_format == "parquet":
            df.to_parquet(save)
        elif save_format == "csv":
            df.to_csv(save, index=False)
        else:
            df.to_json(save, orient="records")
        click.echo(f"saved {len(df)} rows to {save}")
    else:
        click.echo(df.to_string(index=False))


@jobs.command("cancel")
@click.argument("job_id")
def jobs_cancel(job_id: str):
    click.echo(json.dumps(_client().cancel_job(job_id)))


@jobs.command("attach")
@click.argument("job_id", required=False)
@click.option("--latest", is_flag=True, help="attach to the most recent job")
def jobs_attach(job_id: Optional[str], latest: bool):
    so = _client()
    if latest or not job_id:
        jobs_df = so.list_jobs()
        if len(jobs_df) == 0:
            click.echo("no jobs")
            return
        job_id = jobs_df.iloc[0]["job_id"]
    so.attach(job_id)


@cli.group()
def datasets():
    """Dataset operations."""


@datasets.command("create")
def datasets_create():
    click.echo(_client().create_dataset())


@datasets.command("list")
def datasets_list():
    for d in _client().list_datasets():
        click.echo(json.dumps(d))


@datasets.command("files")
@click.argument("dataset_id")
def datasets_files(dataset_id: str):
    for f in _client().list_dataset_files(dataset_id):
        click.echo(f)


@datasets.command("upload")
@click.argument("dataset_id")
@click.argument("paths", nargs=-1, type=click.Path(exists=True))
def datasets_upload(dataset_id: str, paths):
    _client().upload_to_dataset(dataset_id, list(paths))


@datasets.command("download")
@click.argument("dataset_id")
@click.option("--file-name", default=None)
@click.option("--output-path", type=click.Path(), default=".")
def datasets_download(dataset_id: str, file_name, output_path):
    names = _client().download_from_dataset(dataset_id, file_name, output_path)
    click.echo(f"downloaded {len(names)} file(s)")


@cli.group()
def cache():
    """Local results-cache operations."""


@cache.command("clear")
def cache_clear():
    n = _client()._clear_job_results_cache()
    click.echo(f"removed {n} cached result file(s)")


@cache.command("show")
def cache_show():
    for item in _client()._show_cache_contents():
        click.echo(json.dumps(item))


@cli.command()
def quotas():
    for i, q in enumerate(_client().get_quotas()):
        click.echo(f"p{i}: rows={q['row_quota']:,} tokens={q['token_quota']:,}")


@cli.command()
def models():
    """List servable models."""
    from .common import list_models

    for m in list_models():
        click.echo(m)


@cli.command()
@click.option("--host", default="127.0.0.1")
@click.option("--port", default=8000, type=int)
@click.option("--device", default="auto")
def serve(host: str, port: int, device: str):
    """Serve the HTTP API (the same endpoint contract as api.sutro.sh)."""
    from .service.http_api import run_server

    run_server(host=host, port=port, device=device)


@cli.command()
def docs():
    click.echo("see README.md and SURVEY.md in the repository")


if __name__ == "__main__":
    cli()
