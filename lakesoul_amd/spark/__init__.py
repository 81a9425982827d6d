"""PySpark adapter (reference: python/src/lakesoul/spark/tables.py).

pyspark is not installed in this build image; with pyspark present the
intended surface is:

    from lakesoul_amd.spark import LakeSoulTable as SparkLakeSoulTable
    SparkLakeSoulTable.for_name(spark, "t1").upsert(df)

backed by arrow conversion (spark df <-> arrow <-> native engine). The
JVM-native Spark DataSource V2 connector is a round-2 item bound to the
C ABI (csrc/capi/)."""


def _require_pyspark():
    try:
        import pyspark  # noqa: F401
    except ImportError as e:
        raise ImportError(
            "pyspark is not installed in this environment; "
            "lakesoul_amd.spark needs the 'pyspark' package"
        ) from e


class LakeSoulTable:
    @staticmethod
    def for_name(spark, name, namespace="default"):
        _require_pyspark()
        from ..tables.catalog import LakeSoulCatalog

        return _SparkTable(spark, LakeSoulCatalog().table(name, namespace))


class _SparkTable:
    def __init__(self, spark, table):
        self.spark = spark
        self.table = table

    def to_df(self):
        return self.spark.createDataFrame(self.table.to_pandas())

    def upsert(self, df):
        self.table.upsert(df.toPandas())
