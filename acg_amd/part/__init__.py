from .partition import partition_rows, read_partition_file, write_partition_file  # noqa: F401
from .subdomain import LocalSystem, HaloPattern, extract_subdomains  # noqa: F401
