#!/usr/bin/env python3
"""Regenerates the golden fixtures in this directory.

The fixtures are the reference's own TPC-H SF1 known-answer vectors
(exact-decimal result sets, produced by the reference's product-test suite):

  qNN_sf1.result <- /root/reference/presto-product-tests/src/main/resources/
                    sql-tests/testcases/hive_tpch/qNN.result
  for NN in 01, 03, 04, 05, 06, 07

They pin (a) the TPC-H dbgen restatement in oracle/tpchgen.c and (b) the
decimal aggregate semantics of oracle/oracle.c — see tests/test_oracle.py.
This script only runs in the build container (where /root/reference exists);
the committed fixtures are what travels to the GPU box.
"""
import shutil
import pathlib

REF = pathlib.Path("/root/reference/presto-product-tests/src/main/resources/"
                   "sql-tests/testcases/hive_tpch")
HERE = pathlib.Path(__file__).parent

for q in ("q01", "q02", "q03", "q04", "q05", "q06", "q07", "q08", "q09",
          "q10", "q11", "q12", "q13", "q14", "q15", "q16", "q17", "q18",
          "q19", "q20", "q21", "q22"):
    shutil.copy(REF / f"{q}.result", HERE / f"{q}_sf1.result")
    print(f"wrote {q}_sf1.result")
