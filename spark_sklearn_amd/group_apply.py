"""Grouped-apply primitive ``gapply``.

Behavioral counterpart of the reference's ``group_apply.py`` ~L20 [R]
(SURVEY.md C7): apply a ``pandas.DataFrame -> pandas.DataFrame`` function
per group.  The reference implements this over a pyspark ``GroupedData``
with a collect_list shuffle + Python UDF; single-node there is no shuffle,
so this is a thin host-side utility kept for API parity (LOW priority per
the survey) -- the oracle it is tested against is
``pandas.groupby().apply`` itself.
"""

from __future__ import annotations

from typing import Callable, Optional, Sequence

import pandas as pd

__all__ = ["gapply"]


def gapply(grouped_data, func: Callable, schema: Optional[Sequence[str]] = None,
           *cols: str, retainGroupColumns: bool = True) -> pd.DataFrame:
    """Apply ``func`` to each group and concatenate the results.

    Parameters
    ----------
    grouped_data : pandas.core.groupby.DataFrameGroupBy or (DataFrame, keys)
        A pandas groupby object, or a ``(df, key_columns)`` pair.
    func : callable(key, pandas.DataFrame) -> pandas.DataFrame
        Called once per group with the group key tuple and the group's rows
        restricted to ``cols`` (all non-key columns when ``cols`` is empty),
        in an unspecified group order; must return a DataFrame conforming to
        ``schema``.
    schema : sequence of column names or None
        Expected output columns.  ``None`` infers from the first non-empty
        result.  A result not matching the schema raises ``ValueError``.
    *cols : str
        Columns passed through to ``func``.
    retainGroupColumns : bool
        Prepend the group key columns to the output (the reference honors
        the ``spark.sql.retainGroupColumns`` conf the same way).
    """
    if isinstance(grouped_data, tuple):
        df, keys = grouped_data
        keys = [keys] if isinstance(keys, str) else list(keys)
        grouped_data = df.groupby(keys, sort=True)
    keys = list(grouped_data.keys) if isinstance(grouped_data.keys, list) \
        else [grouped_data.keys]

    out_frames = []
    for key, group in grouped_data:
        if not isinstance(key, tuple):
            key = (key,)
        sub = group[list(cols)] if cols else group.drop(columns=keys)
        result = func(key, sub.reset_index(drop=True))
        if result is None or len(result) == 0:
            continue
        if not isinstance(result, pd.DataFrame):
            raise ValueError(
                f"gapply func must return a pandas DataFrame, got "
                f"{type(result).__name__}")
        if schema is not None:
            if list(result.columns) != list(schema):
                raise ValueError(
                    f"gapply result columns {list(result.columns)} do not "
                    f"match schema {list(schema)}")
        elif out_frames:
            if list(result.columns) != list(out_frames[0].columns[len(keys):]
                                            if retainGroupColumns
                                            else out_frames[0].columns):
                raise ValueError("gapply results have inconsistent columns")
        result = result.reset_index(drop=True)
        if retainGroupColumns:
            for i, k in enumerate(keys):
                result.insert(i, k, [key[i]] * len(result))
        out_frames.append(result)

    if not out_frames:
        columns = (keys if retainGroupColumns else []) + \
            (list(schema) if schema is not None else [])
        return pd.DataFrame(columns=columns)
    return pd.concat(out_frames, ignore_index=True)
