"""Benchmark workloads (BASELINE.json configs): TeraSort, PageRank,
SQL sort-merge join, groupByKey. The reference validated against real
Spark workloads (README.md:7-31); these are their standalone analogs."""
