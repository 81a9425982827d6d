"""Console CLI (reference: rust/lakesoul-console + lakesoul-datafusion cli.rs).

    python -m lakesoul_amd.cli list-tables
    python -m lakesoul_amd.cli describe t1
    python -m lakesoul_amd.cli scan t1 --limit 10 --filters "gt(id, 5)"
    python -m lakesoul_amd.cli write t1 data.parquet
    python -m lakesoul_amd.cli compact t1
    python -m lakesoul_amd.cli history t1
"""

from __future__ import annotations

import json

import click


def _catalog():
    from .tables.catalog import LakeSoulCatalog

    return LakeSoulCatalog()


@click.group()
def cli():
    """lakesoul_amd — MI355X-native lakehouse engine."""


@cli.command("sql")
@click.argument("query", required=False)
@click.option("--device", default=None, help="scan device (e.g. cuda:0)")
@click.option("-f", "--file", "script", type=click.Path(exists=True),
              help="run statements from a .sql file (';'-separated)")
def sql(query, device, script):
    """Run one SQL statement, a .sql script, or start the interactive
    console (reference: rust/lakesoul-console)."""
    from .sql import execute_sql, repl

    cat = _catalog()
    if script:
        with open(script) as fh:
            text = fh.read()
        for stmt in [x.strip() for x in text.split(";") if x.strip()]:
            df = execute_sql(cat, stmt, device=device)
            click.echo(df.to_string(index=False))
        return
    if query:
        df = execute_sql(cat, query, device=device)
        try:
            from tabulate import tabulate

            click.echo(tabulate(df, headers="keys", tablefmt="psql", showindex=False))
        except ImportError:
            click.echo(df.to_string(index=False))
    else:
        repl(cat, device=device)


@cli.command("list-tables")
@click.option("--namespace", default="default")
def list_tables(namespace):
    for t in _catalog().list_tables(namespace):
        click.echo(t)


@cli.command("create-table")
@click.argument("name")
@click.option("--schema", required=True, help='e.g. "id:int64,v:float64,s:string"')
@click.option("--primary-keys", default="", help="comma-separated")
@click.option("--range-partitions", default="")
@click.option("--hash-bucket-num", default=1, type=int)
@click.option("--namespace", default="default")
def create_table(name, schema, primary_keys, range_partitions, hash_bucket_num, namespace):
    from .io.schema import Field, Schema

    fields = []
    for part in schema.split(","):
        fname, dtype = part.split(":")
        fields.append(Field(fname.strip(), dtype.strip()))
    t = _catalog().create_table(
        name,
        Schema(fields),
        primary_keys=[c for c in primary_keys.split(",") if c],
        range_partitions=[c for c in range_partitions.split(",") if c],
        hash_bucket_num=hash_bucket_num,
        namespace=namespace,
    )
    click.echo(f"created {t.table_id} at {t.table_path}")


@cli.command()
@click.argument("name")
@click.option("--namespace", default="default")
def describe(name, namespace):
    t = _catalog().table(name, namespace)
    click.echo(json.dumps({
        "table_id": t.table_id,
        "path": t.table_path,
        "schema": [(f.name, f.dtype, f.nullable) for f in t.schema],
        "primary_keys": t.primary_keys,
        "range_partitions": t.range_keys,
        "hash_bucket_num": t.hash_bucket_num,
        "partitions": t.partition_descs(),
    }, indent=1))


@cli.command()
@click.argument("name")
@click.option("--limit", default=20, type=int)
@click.option("--columns", default=None)
@click.option("--filters", default=None, help="filter DSL, e.g. gt(id, 5)")
@click.option("--version", default=None, type=int)
@click.option("--namespace", default="default")
def scan(name, limit, columns, filters, version, namespace):
    t = _catalog().table(name, namespace)
    tbl = t.scan(
        columns=columns.split(",") if columns else None,
        filters=filters,
        version=version,
    ).to_arrow()
    click.echo(tbl.slice(0, limit).to_pandas().to_string())
    click.echo(f"-- {tbl.num_rows} rows")


@cli.command()
@click.argument("name")
@click.argument("parquet_file")
@click.option("--namespace", default="default")
def write(name, parquet_file, namespace):
    import pyarrow.parquet as pq

    t = _catalog().table(name, namespace)
    tbl = pq.read_table(parquet_file)
    t.write(tbl)
    click.echo(f"wrote {tbl.num_rows} rows")


@cli.command()
@click.argument("name")
@click.option("--partition", default=None)
@click.option("--namespace", default="default")
def compact(name, partition, namespace):
    t = _catalog().table(name, namespace)
    t.compaction(partition)
    click.echo("compaction done")


@cli.command()
@click.argument("name")
@click.option("--namespace", default="default")
def history(name, namespace):
    t = _catalog().table(name, namespace)
    for desc in t.partition_descs():
        cur = t.client.store.get_latest_partition_info(t.table_id, desc)
        for p in t.client.store.get_partition_versions_in_range(t.table_id, desc, 0, cur.version):
            click.echo(f"{desc} v{p.version} {p.commit_op.name} ts={p.timestamp} commits={len(p.snapshot)}")


@cli.command("drop-table")
@click.argument("name")
@click.option("--namespace", default="default")
@click.option("--delete-data", is_flag=True)
def drop_table(name, namespace, delete_data):
    _catalog().drop_table(name, namespace, delete_data=delete_data)
    click.echo("dropped")


if __name__ == "__main__":
    cli()
