"""DataSet — drop-in surface of the reference's tuplex/python/tuplex/dataset.py.

Operator chaining builds the logical op list consumed by plan.build_stage; the same
method names, argument shapes and semantics (map dataset.py:49, filter :83,
collect :113, take :125, resolve :162, withColumn :201, mapColumn :231,
selectColumns :262, renameColumn :293, ignore :319, aggregate :593, tocsv :500).
"""
from typing import Any, Callable, List, Optional, Union


class DataSet:
    def __init__(self, context, source, ops=None):
        self._context = context
        self._source = source          # ParallelizeSource | CsvSource
        self._ops = ops or []
        self._last_outcome = None

    def _chain(self, op) -> "DataSet":
        return DataSet(self._context, self._source, self._ops + [op])

    # ---- transformations ----------------------------------------------------
    def map(self, ftor: Callable) -> "DataSet":
        assert callable(ftor), "need a function"
        return self._chain(("map", ftor))

    def filter(self, ftor: Callable) -> "DataSet":
        assert callable(ftor), "need a function"
        return self._chain(("filter", ftor))

    def withColumn(self, column: str, ftor: Callable) -> "DataSet":
        assert isinstance(column, str) and callable(ftor)
        return self._chain(("withColumn", column, ftor))

    def mapColumn(self, column: Union[int, str], ftor: Callable) -> "DataSet":
        assert callable(ftor)
        return self._chain(("mapColumn", column, ftor))

    def selectColumns(self, columns: List[Union[str, int]]) -> "DataSet":
        if not isinstance(columns, list):
            columns = [columns]
        return self._chain(("selectColumns", columns))

    def renameColumn(self, key, newColumnName: str) -> "DataSet":
        return self._chain(("renameColumn", key, newColumnName))

    def resolve(self, eclass, ftor: Callable) -> "DataSet":
        assert callable(ftor)
        return self._chain(("resolve", eclass, ftor))

    def ignore(self, eclass) -> "DataSet":
        return self._chain(("ignore", eclass))

    def aggregate(self, combine: Callable, aggregate: Callable,
                  initial_value: Any) -> "DataSet":
        return self._chain(("aggregate", combine, aggregate, initial_value))

    def aggregateByKey(self, combine, aggregate, initial_value,
                       key_columns) -> "DataSet":
        if not isinstance(key_columns, list):
            key_columns = [key_columns]
        return self._chain(("aggregateByKey", combine, aggregate, initial_value,
                            key_columns))

    def unique(self) -> "DataSet":
        """Distinct rows (hashtable sink like aggregateByKey; output order is
        unpinned, as the reference's hashmap iteration order is)."""
        return self._chain(("unique",))

    def _join(self, dsRight, leftKeyColumn, rightKeyColumn, prefixes, suffixes,
              how):
        """Inner/left join, right side materialized as the hash-BUILD side
        (reference: logical/JoinOperator.cc; output layout JoinOperator.cc:164:
        | left cols except key | key (left name) | right cols except key |;
        HashJoinStage probe — SURVEY.md §8f-3). This round the build side needs
        UNIQUE non-null keys (dimension-table joins); duplicate keys raise."""
        lp = ls = rp = rs = ""
        if prefixes:
            p = tuple(prefixes)
            lp, rp = p[0] or "", p[1] or ""
        if suffixes:
            sfx = tuple(suffixes)
            ls, rs = sfx[0] or "", sfx[1] or ""
        rcols = dsRight.columns
        if not rcols:
            raise ValueError("join: right dataset needs named columns")
        if not dsRight._ops and getattr(dsRight._source, "kind", "") == "mem":
            rrows = list(dsRight._source.data)  # no pipeline: direct
        else:
            rrows = dsRight.collect()
        rrows = [(v if isinstance(v, tuple) else (v,)) for v in rrows]
        return self._chain(("join", rrows, list(rcols), leftKeyColumn,
                            rightKeyColumn, how, lp, ls, rp, rs))

    def join(self, dsRight, leftKeyColumn: str, rightKeyColumn: str,
             prefixes=None, suffixes=None) -> "DataSet":
        return self._join(dsRight, leftKeyColumn, rightKeyColumn, prefixes,
                          suffixes, "inner")

    def leftJoin(self, dsRight, leftKeyColumn: str, rightKeyColumn: str,
                 prefixes=None, suffixes=None) -> "DataSet":
        return self._join(dsRight, leftKeyColumn, rightKeyColumn, prefixes,
                          suffixes, "left")

    def cache(self, store_specialized: bool = True) -> "DataSet":
        """Materialize the pipeline so far; downstream operators start from
        the materialized rows instead of re-executing (CacheOperator,
        core/src/logical/CacheOperator.cc — SURVEY.md §8f-4). Exception rows
        are STORED at the cache point, not resolved — resolvers attached
        after cache() replay them through the original op chain, matching
        the reference's deferred resolution. Pipelines ending in
        aggregate/unique consume row identity, so those cache the final
        values (nothing left to resolve into)."""
        from .context import CachedSource, ParallelizeSource
        has_agg = any(op[0] in ("aggregate", "aggregateByKey", "unique")
                      for op in self._ops)
        outcome = self._context._execute(self, keep_exceptions=not has_agg)
        if (has_agg or outcome.row_keys is None
                or outcome.pending_replayer is None):
            src = ParallelizeSource(list(outcome.rows), self.columns)
        else:
            src = CachedSource(list(outcome.rows), outcome.row_keys,
                               outcome.pending, outcome.pending_replayer,
                               list(self._ops), self.columns)
        ds = DataSet(self._context, src)
        ds._last_outcome = outcome
        return ds

    # ---- actions ------------------------------------------------------------
    def collect(self) -> List[Any]:
        outcome = self._context._execute(self)
        self._last_outcome = outcome
        return outcome.rows

    def take(self, nrows: int = 5) -> List[Any]:
        return self.collect()[:nrows]

    def show(self, nrows: Optional[int] = None) -> None:
        rows = self.collect()
        if nrows is not None:
            rows = rows[:nrows]
        for r in rows:
            print(r)

    def tocsv(self, path: str, **kw) -> None:
        outcome = self._context._execute(self, sink=("csv", path))
        self._last_outcome = outcome

    def toorc(self, path: str, **kw) -> None:
        """write output to ORC (io OrcWriter analog; Arrow writer on host —
        the GPU produces the rows, pyarrow serializes the columnar file)."""
        import pyarrow as pa
        import pyarrow.orc as paorc
        outcome = self._context._execute(self)
        self._last_outcome = outcome
        rows = [(v if isinstance(v, tuple) else (v,)) for v in outcome.rows]
        ncols = len(rows[0]) if rows else len(self.columns or []) or 1
        names = self.columns or ["column%d" % i for i in range(ncols)]
        cols = {names[k]: [r[k] for r in rows] for k in range(ncols)}
        paorc.write_table(pa.table(cols), path)

    # ---- introspection ------------------------------------------------------
    @property
    def columns(self) -> Optional[List[str]]:
        cols = self._source.columns
        for op in self._ops:
            if op[0] == "map":
                cols = None
            elif op[0] == "withColumn":
                if cols and op[1] not in cols:
                    cols = cols + [op[1]]
            elif op[0] == "selectColumns":
                cols = [c if isinstance(c, str) else (cols[c] if cols else None)
                        for c in op[1]]
            elif op[0] == "renameColumn":
                if cols:
                    cols = [op[2] if c == op[1] else c for c in cols]
            elif op[0] == "join":
                _, _rr, rcols, lk, rk, _how, lp, ls, rp, rs = op
                if cols:
                    cols = ([lp + c + ls for c in cols if c != lk] +
                            [lp + lk + ls] +
                            [rp + c + rs for c in rcols if c != rk])
        return cols

    @property
    def exception_counts(self) -> dict:
        if self._last_outcome is None:
            return {}
        return dict(self._last_outcome.exception_counts)
