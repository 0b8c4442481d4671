"""NDS (TPC-DS-derived) q1-q99 harness.

The reference's headline comparison is the NDS benchmark run by the
RAPIDS Accelerator plugin on top of spark-rapids-jni (SURVEY.md §6,
BASELINE.json configs[4]). The reference repo itself contains no query
runner — Spark is the engine; this package is the MI355X-native analog of
that engine slice: a tiny relational plan IR (scan/filter/project/join/
aggregate/window/sort/limit/exchange), executed either on GPU through the
hand-written HIP kernels (ops/join.py, ops/aggregate.py, ...) or on CPU
through plain torch reference implementations used as the correctness
oracle, plus synthetic TPC-DS-shaped generators and a power-run driver.
"""
from .expr import Col, Lit, col, lit, date_lit, case_when, coalesce, is_null
from .plan import (Scan, Filter, Project, Join, Agg, Window, Sort, Limit,
                   Union, Distinct, Engine)
from .schema import gen_catalog, TABLES
