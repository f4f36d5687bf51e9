"""L8 observability: log parsing, plotting, live monitoring, stats mesh.

Parity surface: /root/reference/utils/{plotting,monitoring,realtime_plotting}.py
and /root/reference/{stats_server,stats_client}.py.
"""
from .log_parse import LogRecord, parse_log_file, parse_log_line  # noqa: F401
