from .csv_ingest import CsvIngest  # noqa: F401
from . import synthetic  # noqa: F401
