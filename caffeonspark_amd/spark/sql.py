"""Minimal Spark-SQL-shaped DataFrame over pandas.

The reference driver materialises feature/validation rows into Spark
DataFrames and writes them as json/parquet (CaffeOnSpark.scala:344-357,
474-506; -outputFormat).  This provides the subset the CaffeOnSpark
driver and the ML-pipeline example use: Row, createDataFrame, select,
collect, count, toPandas, write.json/parquet.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional


class Row(dict):
    """Spark Row: field access by attribute or key."""

    def __getattr__(self, name):
        try:
            return self[name]
        except KeyError:
            raise AttributeError(name)

    def asDict(self) -> Dict[str, Any]:
        return dict(self)


class DataFrameWriter:
    def __init__(self, df: "DataFrame"):
        self._df = df
        self._mode = "overwrite"

    def mode(self, m: str) -> "DataFrameWriter":
        self._mode = m
        return self

    def json(self, path: str) -> None:
        self._df.toPandas().to_json(path, orient="records", lines=True)

    def parquet(self, path: str) -> None:
        self._df.toPandas().to_parquet(path)

    def format(self, fmt: str):
        self._fmt = fmt
        return self

    def save(self, path: str) -> None:
        getattr(self, getattr(self, "_fmt", "json"))(path)


class DataFrame:
    def __init__(self, rows: List[Row], columns: Optional[List[str]] = None):
        self._rows = [r if isinstance(r, Row) else Row(r) for r in rows]
        if columns is None:
            columns = list(self._rows[0].keys()) if self._rows else []
        self.columns = columns

    def collect(self) -> List[Row]:
        return list(self._rows)

    def count(self) -> int:
        return len(self._rows)

    def select(self, *cols) -> "DataFrame":
        cols = [c for c in cols]
        return DataFrame([Row({c: r.get(c) for c in cols})
                          for r in self._rows], cols)

    def toPandas(self):
        import pandas as pd
        return pd.DataFrame([dict(r) for r in self._rows],
                            columns=self.columns or None)

    def show(self, n: int = 20, truncate: bool = True) -> None:
        print(self.toPandas().head(n))

    @property
    def write(self) -> DataFrameWriter:
        return DataFrameWriter(self)

    def __iter__(self):
        return iter(self._rows)


class SQLContext:
    def __init__(self, sc):
        self._sc = sc

    def createDataFrame(self, rows, schema: Optional[List[str]] = None
                        ) -> DataFrame:
        rows = list(rows)
        if schema and rows and not isinstance(rows[0], dict):
            rows = [Row(zip(schema, r)) for r in rows]
        return DataFrame(rows, schema)

    def read_parquet(self, path: str) -> DataFrame:
        import pandas as pd
        pdf = pd.read_parquet(path)
        return DataFrame([Row(rec) for rec in pdf.to_dict("records")],
                         list(pdf.columns))
