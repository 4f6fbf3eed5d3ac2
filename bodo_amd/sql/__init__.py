"""bodo_amd.sql: SQL frontend (reference: BodoSQL BodoSQLContext,
BodoSQL/bodosql/context.py:504) — parses SQL natively (no JVM/Calcite) and
executes it on the same logical-plan/columnar-HIP engine as the pandas API.
"""

from __future__ import annotations

from typing import Dict


class BodoSQLContext:
    def __init__(self, tables: Dict[str, object]):
        from ..pandas.frame import BodoDataFrame, from_pandas_df

        import pandas as pd

        self.tables = {}
        for name, t in tables.items():
            if isinstance(t, pd.DataFrame):
                t = from_pandas_df(t)
            self.tables[name] = t

    def add_or_replace_view(self, name: str, table) -> "BodoSQLContext":
        new = dict(self.tables)
        new[name] = table
        return BodoSQLContext(new)

    def remove_view(self, name: str) -> "BodoSQLContext":
        new = dict(self.tables)
        new.pop(name, None)
        return BodoSQLContext(new)

    _plan_cache = {}  # (reference: BodoSqlPlanCache, bodo/sql_plan_cache.py)

    def sql(self, query: str):
        from ..pandas.frame import BodoDataFrame
        from .parser import Parser
        from .planner import Planner

        stripped = query.strip()
        low = stripped.lower()
        if low.startswith("explain"):
            from ..engine.optimizer import optimize
            from ..plan.nodes import explain as _explain

            inner = self.sql(stripped[len("explain"):])
            return _explain(optimize(inner._plan))
        if low.startswith("create table") or low.startswith(
                "create or replace table"):
            # CTAS: CREATE [OR REPLACE] TABLE name AS SELECT ...
            # (reference: BodoSQL DDL executed directly, context.py:504)
            import re as _re

            m = _re.match(r"create\s+(?:or\s+replace\s+)?table\s+(\w+)\s+as"
                          r"\s+(.*)$", stripped,
                          _re.IGNORECASE | _re.DOTALL)
            if not m:
                raise ValueError(f"unsupported DDL: {stripped[:60]}")
            name, select = m.group(1), m.group(2)
            self.tables[name.lower()] = self.sql(select)
            BodoSQLContext._plan_cache.clear()
            return None
        if low.startswith("insert into"):
            # INSERT INTO name SELECT ... | INSERT INTO name VALUES (...)
            import re as _re

            m = _re.match(r"insert\s+into\s+(\w+)\s+(.*)$", stripped,
                          _re.IGNORECASE | _re.DOTALL)
            if not m:
                raise ValueError(f"unsupported INSERT: {stripped[:60]}")
            name, rest = m.group(1).lower(), m.group(2).strip()
            if name not in self.tables:
                raise KeyError(f"unknown table {name}")
            base = self.tables[name]
            if rest.lower().startswith("values"):
                import ast as _pyast

                import pandas as _pd

                rows = _pyast.literal_eval(
                    "[" + rest[len("values"):].strip() + "]")
                new = _pd.DataFrame(rows, columns=list(base.columns))
                from ..pandas import from_pandas as _fp

                add = _fp(new)
            else:
                add = self.sql(rest)
            from ..pandas import concat as _concat

            self.tables[name] = _concat([base, add])
            BodoSQLContext._plan_cache.clear()
            return None
        if low.startswith("delete from"):
            # DELETE FROM name [WHERE cond] — keep the complement
            import re as _re

            m = _re.match(r"delete\s+from\s+(\w+)(?:\s+where\s+(.*))?$",
                          stripped, _re.IGNORECASE | _re.DOTALL)
            if not m:
                raise ValueError(f"unsupported DELETE: {stripped[:60]}")
            name, cond = m.group(1).lower(), m.group(2)
            if name not in self.tables:
                raise KeyError(f"unknown table {name}")
            if cond is None:
                self.tables[name] = self.tables[name].head(0)
            else:
                self.tables[name] = self.sql(
                    f"select * from {name} where not ({cond})")
            BodoSQLContext._plan_cache.clear()
            return None
        if low.startswith("update "):
            # UPDATE name SET c1 = e1 [, ...] [WHERE cond]
            import re as _re

            m = _re.match(r"update\s+(\w+)\s+set\s+(.*?)"
                          r"(?:\s+where\s+(.*))?$",
                          stripped, _re.IGNORECASE | _re.DOTALL)
            if not m:
                raise ValueError(f"unsupported UPDATE: {stripped[:60]}")
            name, sets, cond = m.group(1).lower(), m.group(2), m.group(3)
            if name not in self.tables:
                raise KeyError(f"unknown table {name}")
            assigns = {}
            for part in _split_top_level_commas(sets):
                col, _, expr = part.partition("=")
                assigns[col.strip().lower()] = expr.strip()
            cols = []
            for c in self.tables[name].columns:
                cl = str(c).lower()
                if cl in assigns:
                    e = assigns[cl]
                    if cond is not None:
                        cols.append(f"case when {cond} then ({e}) "
                                    f"else {c} end as {c}")
                    else:
                        cols.append(f"({e}) as {c}")
                else:
                    cols.append(str(c))
            self.tables[name] = self.sql(
                f"select {', '.join(cols)} from {name}")
            BodoSQLContext._plan_cache.clear()
            return None
        key = (query, tuple(sorted((n, id(t._lazy_plan))
                                   for n, t in self.tables.items())))
        hit = BodoSQLContext._plan_cache.get(key)
        if hit is None:
            q = Parser(query).parse()
            hit = Planner(self.tables).plan(q)
            if len(BodoSQLContext._plan_cache) > 256:
                BodoSQLContext._plan_cache.clear()
            BodoSQLContext._plan_cache[key] = hit
        plan, names = hit
        return BodoDataFrame(plan, names)

    # convenience parity alias
    def convert_to_pandas(self, query: str) -> str:  # pragma: no cover
        raise NotImplementedError(
            "bodo_amd executes SQL directly on the C++ backend path; "
            "pandas-codegen output is not produced")


def _split_top_level_commas(s: str):
    """Split on commas not inside parentheses or quotes."""
    parts, depth, cur, q = [], 0, [], None
    for ch in s:
        if q:
            cur.append(ch)
            if ch == q:
                q = None
            continue
        if ch in ("'", '"'):
            q = ch
            cur.append(ch)
        elif ch == "(":
            depth += 1
            cur.append(ch)
        elif ch == ")":
            depth -= 1
            cur.append(ch)
        elif ch == "," and depth == 0:
            parts.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    if cur:
        parts.append("".join(cur))
    return parts
