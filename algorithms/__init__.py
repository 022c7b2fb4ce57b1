"""Built-in algorithm packages.

The MI355X analog of vantage6 algorithm Docker images: ordinary Python
modules obeying the I/O contract, dispatched by
name via ``vantage6_amd.algorithm.wrap``. A task's ``image`` field holds the
module path (e.g. ``algorithms.column_mean``).
"""
